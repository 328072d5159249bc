"""Matrix printers (reference: raft/matrix/print.hpp)."""
from __future__ import annotations

import torch


def print_matrix(x: torch.Tensor, name: str = "", max_rows: int = 8,
                 max_cols: int = 8, file=None) -> str:
    """Human-readable preview of a (possibly device) matrix."""
    import io
    out = io.StringIO()
    r, c = (x.shape + (1,))[:2] if x.dim() >= 2 else (x.shape[0], 1)
    hdr = f"{name or 'matrix'} [{tuple(x.shape)} {x.dtype} {x.device}]"
    print(hdr, file=out)
    view = x.detach()
    if view.dim() == 1:
        view = view.unsqueeze(0)
    rr = min(max_rows, view.shape[0])
    cc = min(max_cols, view.shape[1])
    host = view[:rr, :cc].cpu()
    for i in range(rr):
        row = " ".join(f"{float(v):10.4g}" for v in host[i])
        more = " ..." if view.shape[1] > cc else ""
        print(f"  {row}{more}", file=out)
    if view.shape[0] > rr:
        print("  ...", file=out)
    s = out.getvalue()
    print(s, end="", file=file)
    return s
