from .lap import linear_assignment, LinearAssignmentProblem

__all__ = ["linear_assignment", "LinearAssignmentProblem"]
