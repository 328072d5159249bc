from .lap import linear_assignment, linear_assignment_batched, LinearAssignmentProblem

__all__ = ["linear_assignment", "linear_assignment_batched", "LinearAssignmentProblem"]
