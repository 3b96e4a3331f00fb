"""Typed pub/sub message broker (capability parity with reference
src/modalities/logging_broker/ — message_broker.py, publisher.py,
messages.py)."""

from dataclasses import dataclass
from enum import Enum
from typing import Generic, TypeVar

T = TypeVar("T")


class MessageTypes(str, Enum):
    HIGH_LEVEL_PROGRESS_UPDATE = "HIGH_LEVEL_PROGRESS_UPDATE"
    BATCH_PROGRESS_UPDATE = "BATCH_PROGRESS_UPDATE"
    ERROR_MESSAGE = "ERROR_MESSAGE"
    EVALUATION_RESULT = "EVALUATION_RESULT"


class ExperimentStatus(str, Enum):
    TRAIN = "TRAIN"
    EVALUATION = "EVALUATION"
    CHECKPOINTING = "CHECKPOINTING"


@dataclass
class ProgressUpdate:
    num_steps_done: int
    experiment_status: ExperimentStatus
    dataloader_tag: str = ""


@dataclass
class Message(Generic[T]):
    message_type: MessageTypes
    payload: T
    global_rank: int = 0
    local_rank: int = 0


class MessageSubscriberIF(Generic[T]):
    def consume_message(self, message: Message[T]) -> None:
        raise NotImplementedError

    def consume_dict(self, message_dict: dict) -> None:
        raise NotImplementedError


class MessageBroker:
    """Routes published messages to subscribers by message type."""

    def __init__(self):
        self._subscriptions: dict[MessageTypes, list[MessageSubscriberIF]] = {}

    def add_subscriber(self, subscription: MessageTypes,
                       subscriber: MessageSubscriberIF) -> None:
        self._subscriptions.setdefault(subscription, []).append(subscriber)

    def distribute_message(self, message: Message) -> None:
        for subscriber in self._subscriptions.get(message.message_type, []):
            subscriber.consume_message(message)


class MessagePublisher(Generic[T]):
    def __init__(self, message_broker: MessageBroker, global_rank: int, local_rank: int):
        self._broker = message_broker
        self._global_rank = global_rank
        self._local_rank = local_rank

    def publish_message(self, payload: T, message_type: MessageTypes) -> None:
        self._broker.distribute_message(Message(
            message_type=message_type, payload=payload,
            global_rank=self._global_rank, local_rank=self._local_rank))
