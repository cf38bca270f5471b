"""Retry with exponential backoff and jitter
(reference src/core/resilience/patterns.py:403-462)."""

from __future__ import annotations

import functools
import random
import time
from typing import Callable, Type


def retry_with_backoff(
    max_attempts: int = 3,
    base_delay: float = 0.1,
    max_delay: float = 5.0,
    jitter: float = 0.1,
    exceptions: tuple[Type[BaseException], ...] = (Exception,),
) -> Callable:
    def deco(fn: Callable) -> Callable:
        @functools.wraps(fn)
        def wrapper(*args, **kwargs):
            last: BaseException | None = None
            for attempt in range(max_attempts):
                try:
                    return fn(*args, **kwargs)
                except exceptions as exc:
                    last = exc
                    if attempt == max_attempts - 1:
                        break
                    delay = min(base_delay * (2 ** attempt), max_delay)
                    delay += random.uniform(0.0, jitter * delay)
                    time.sleep(delay)
            raise last  # type: ignore[misc]

        return wrapper

    return deco
