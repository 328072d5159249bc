"""Packaging for raft_amd.

The native extension is built in-tree by build_ext.py (hipcc, gfx950);
`python setup.py build_ext --inplace` delegates there so the conventional
command works. The .so stays in-tree (raft_amd/_C*.so) — NOT installed into
site-packages — so repo snapshots carry it (see __graft_entry__.build).
"""
import sys

from setuptools import Command, find_packages, setup


class BuildExtInTree(Command):
    user_options = [("inplace", "i", "build in-tree (always true here)")]

    def initialize_options(self):
        self.inplace = True

    def finalize_options(self):
        pass

    def run(self):
        import build_ext

        build_ext.build()


setup(
    name="raft_amd",
    version="0.1.0",
    description="MI355X-native (CDNA4/gfx950) ML and data-mining primitives "
                "with the capability surface of rapidsai/raft",
    packages=find_packages(include=["raft_amd", "raft_amd.*"]),
    package_data={"raft_amd": ["_C*.so"]},
    python_requires=">=3.10",
    cmdclass={"build_ext": BuildExtInTree},
)
