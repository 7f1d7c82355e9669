from .cel import CostProgram, CostVars, compile_program, evaluate

__all__ = ["CostProgram", "CostVars", "compile_program", "evaluate"]
