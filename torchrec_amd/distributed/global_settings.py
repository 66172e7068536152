"""Module-level toggles (reference: torchrec/distributed/global_settings.py)."""

PROPAGATE_DEVICE: bool = False


def set_propagate_device(val: bool) -> None:
    global PROPAGATE_DEVICE
    PROPAGATE_DEVICE = val


def get_propagate_device() -> bool:
    return PROPAGATE_DEVICE
