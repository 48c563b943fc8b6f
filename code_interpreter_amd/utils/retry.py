"""Small async retry helper (tenacity is not in this image).

Matches the retry shape the reference relies on: N attempts on a given
exception type with jittered exponential backoff
(kubernetes_code_executor.py:75-79, 191-195).
"""

import asyncio
import logging
import random
from typing import Awaitable, Callable, Tuple, Type, TypeVar

T = TypeVar("T")

logger = logging.getLogger("code_executor")


async def async_retry(
    fn: Callable[[], Awaitable[T]],
    *,
    attempts: int = 3,
    retry_on: Tuple[Type[BaseException], ...] = (RuntimeError,),
    min_backoff: float = 4.0,
    max_backoff: float = 10.0,
) -> T:
    last_exc: BaseException | None = None
    for attempt in range(attempts):
        try:
            return await fn()
        except retry_on as e:
            last_exc = e
            if attempt == attempts - 1:
                break
            delay = min(max_backoff, min_backoff * (2**attempt))
            delay *= 0.5 + random.random() / 2
            logger.warning(
                "attempt %d/%d failed (%s); retrying in %.1fs",
                attempt + 1,
                attempts,
                e,
                delay,
            )
            await asyncio.sleep(delay)
    assert last_exc is not None
    raise last_exc
