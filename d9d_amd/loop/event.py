"""Typed pub/sub event bus (reference: d9d/loop/event/core.py:10-71 + catalogue)."""

from contextlib import contextmanager
from dataclasses import dataclass
from typing import Any, Callable, Generic, Iterator, TypeVar

TContext = TypeVar("TContext")


@dataclass(frozen=True)
class Event(Generic[TContext]):
    name: str


class EventBus:
    def __init__(self) -> None:
        self._subscribers: dict[str, list[Callable]] = {}

    def subscribe(self, event: Event, handler: Callable) -> None:
        self._subscribers.setdefault(event.name, []).append(handler)

    def trigger(self, event: Event, context: Any = None) -> None:
        for handler in self._subscribers.get(event.name, []):
            handler(context)

    @contextmanager
    def bounded(self, pre: Event, post: Event, context: Any = None) -> Iterator[None]:
        self.trigger(pre, context)
        try:
            yield
        finally:
            self.trigger(post, context)


# -- train catalogue (reference: loop/event/catalogue/train.py:63-117) --------

TRAIN_CONFIGURE_PRE = Event("train.configure.pre")
TRAIN_CONFIGURE_POST = Event("train.configure.post")
TRAIN_RUN_PRE = Event("train.run.pre")
TRAIN_RUN_POST = Event("train.run.post")
TRAIN_STEP_PRE = Event("train.step.pre")
TRAIN_STEP_POST = Event("train.step.post")
TRAIN_FORWARD_BACKWARD_PRE = Event("train.forward_backward.pre")
TRAIN_FORWARD_BACKWARD_POST = Event("train.forward_backward.post")
TRAIN_OPTIMIZER_STEP_PRE = Event("train.optimizer_step.pre")
TRAIN_OPTIMIZER_STEP_POST = Event("train.optimizer_step.post")
TRAIN_CHECKPOINT_PRE = Event("train.checkpoint.pre")
TRAIN_CHECKPOINT_POST = Event("train.checkpoint.post")
TRAIN_SLEEP_PRE = Event("train.sleep.pre")
TRAIN_SLEEP_POST = Event("train.sleep.post")
TRAIN_WAKE_PRE = Event("train.wake.pre")
TRAIN_WAKE_POST = Event("train.wake.post")

# -- inference catalogue (reference: catalogue/inference.py:25-52) ------------

INFER_CONFIGURE_PRE = Event("infer.configure.pre")
INFER_CONFIGURE_POST = Event("infer.configure.post")
INFER_RUN_PRE = Event("infer.run.pre")
INFER_RUN_POST = Event("infer.run.post")
INFER_BATCH_PRE = Event("infer.batch.pre")
INFER_BATCH_POST = Event("infer.batch.post")
