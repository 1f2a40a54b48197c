"""Module-level amp state (reference: removed apex/amp/_amp_state.py,
surface reconstructed from tests/L1/common/main_amp.py and
examples/dcgan/main_amp.py — see SURVEY.md §0)."""


class AmpState:
    def __init__(self):
        self.initialized = False
        self.opt_properties = None
        self.loss_scalers = []
        self.optimizers = []
        self.models = []
        self.verbosity = 1
        self.cast_dtype = None  # set by initialize; used by half_function

    def reset(self):
        self.__init__()


_amp_state = AmpState()


def maybe_print(msg, rank0_only=True):
    if _amp_state.verbosity > 0:
        try:
            import torch.distributed as dist

            if rank0_only and dist.is_available() and dist.is_initialized() and dist.get_rank() != 0:
                return
        except Exception:
            pass
        print(msg)
