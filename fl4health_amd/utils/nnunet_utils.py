"""nnU-Net glue utilities (capability of reference fl4health/utils/nnunet_utils.py:115-491)."""
from fl4health_amd.models.unet3d import DeepSupervisionLoss, PolyLRScheduler  # noqa: F401


def use_default_signal_handlers(fn):
    """Decorator parity: restore default signal handlers around subprocess-spawning fns."""
    import functools
    import signal

    @functools.wraps(fn)
    def wrapped(*args, **kwargs):
        sigint = signal.getsignal(signal.SIGINT)
        sigterm = signal.getsignal(signal.SIGTERM)
        signal.signal(signal.SIGINT, signal.SIG_DFL)
        signal.signal(signal.SIGTERM, signal.SIG_DFL)
        try:
            return fn(*args, **kwargs)
        finally:
            signal.signal(signal.SIGINT, sigint)
            signal.signal(signal.SIGTERM, sigterm)

    return wrapped
