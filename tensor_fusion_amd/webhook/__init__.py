from .mutator import PodMutator

__all__ = ["PodMutator"]
