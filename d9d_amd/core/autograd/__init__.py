from .grad_context import GradDirection, GlobalGradContext, GLOBAL_GRAD_CONTEXT

__all__ = ["GradDirection", "GlobalGradContext", "GLOBAL_GRAD_CONTEXT"]
