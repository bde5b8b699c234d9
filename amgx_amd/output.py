"""Print-callback indirection (reference src/misc.cu ``amgx_output`` /
``AMGX_register_print_callback``): every library print — per-iteration
residual tables, grid statistics, timings, error messages — goes through
:func:`amgx_output` so a host application's registered callback captures
ALL output, exactly like the reference."""

from __future__ import annotations

from typing import Callable, Optional

print_callback: Optional[Callable[[str], None]] = None


def amgx_output(msg: str) -> None:
    cb = print_callback
    if cb is not None:
        try:
            cb(msg)
            return
        except Exception:   # a broken callback must not kill the solve
            pass
    print(msg, end="")
