"""Native-extension re-exports (mirrors reference persia/prelude.py).

The reference registers the PyO3 ``persia_core`` submodules here; we expose
the HIP extension ``persia_amd._C`` accessors instead."""
from persia_amd.ops import native, native_available  # noqa: F401
from persia_amd.core.queue import (  # noqa: F401
    PersiaBatchDataChannel,
    PersiaBatchDataReceiver,
    PersiaBatchDataSender,
)
