from baton_amd.runtime.local import LocalTrainer

__all__ = ["LocalTrainer"]
