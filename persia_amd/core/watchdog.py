"""Stall/deadlock watchdog.

Mirrors the reference's parking_lot deadlock-detection background thread
(persia-common/src/utils.rs:22-48): opt-in via ``PERSIA_DEADLOCK_DETECTION=1``,
dumps every thread's stack periodically so hung pipelines are diagnosable."""
import os
import sys
import threading
import traceback

from persia_amd.logger import get_default_logger

_logger = get_default_logger("persia_amd.watchdog")

_started = False
_stop = threading.Event()


def stop_deadlock_detection() -> None:
    """Stop the watchdog thread (tests / shutdown)."""
    global _started
    _stop.set()
    _started = False


def maybe_start_deadlock_detection(interval_sec: float = 60.0) -> None:
    global _started
    if _started or os.environ.get("PERSIA_DEADLOCK_DETECTION", "0") not in ("1", "true"):
        return
    _started = True
    _stop.clear()

    def loop():
        while not _stop.wait(interval_sec):
            frames = sys._current_frames()
            lines = [f"--- watchdog: {len(frames)} threads ---"]
            for tid, frame in frames.items():
                name = next(
                    (t.name for t in threading.enumerate() if t.ident == tid), str(tid)
                )
                lines.append(f"thread {name}:")
                lines.extend(l.rstrip() for l in traceback.format_stack(frame))
            _logger.warning("\n".join(lines))

    threading.Thread(target=loop, daemon=True, name="persia-watchdog").start()
