"""train_with_progress — tqdm loop around ps.round().

Reference parity: utils/training.py:7-34.
"""
from __future__ import annotations

from typing import Any, Callable, Optional


async def train_with_progress(
    ps: Any,
    rounds: int,
    *,
    eval_callback: Optional[Callable[[int], dict]] = None,
    eval_interval: int = 10,
    progress: bool = True,
) -> None:
    iterator = range(rounds)
    bar = None
    if progress:
        try:
            from tqdm import tqdm

            bar = tqdm(total=rounds, desc="training")
        except ImportError:
            bar = None
    for r in iterator:
        await ps.round()
        if bar is not None:
            bar.update(1)
            if eval_callback is not None and (r + 1) % eval_interval == 0:
                metrics = eval_callback(r + 1)
                if metrics:
                    bar.set_postfix(metrics)
        elif eval_callback is not None and (r + 1) % eval_interval == 0:
            eval_callback(r + 1)
    if bar is not None:
        bar.close()
