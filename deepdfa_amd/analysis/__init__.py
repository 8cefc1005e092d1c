from .dataflow import CPG, ReachingDefinitions, VariableDefinition, MOD_OPS

__all__ = ["CPG", "ReachingDefinitions", "VariableDefinition", "MOD_OPS"]
