"""In-process message bus replacing the reference's Kafka transport.

The reference moves every inter-stage message over Kafka topics
(config.py:15: vix, volume, cot, ind, deep, predict_timestamp, prediction;
producers at producer.py:130-136, consumer at predict.py:19-30). Here the
same topic semantics run in-process: named FIFO topics with publish /
subscribe / seek-to-end, no brokers.
"""
import threading
from collections import deque
from typing import Any, Callable, Dict, Iterator, List, Optional

from ..config import TOPICS


class Topic:
    def __init__(self, name: str, maxlen: int = 65536):
        self.name = name
        self._buf: deque = deque(maxlen=maxlen)
        self._offset0 = 0  # logical offset of _buf[0]
        self._cv = threading.Condition()
        self._subscribers: List[Callable[[Any], None]] = []

    def publish(self, value: Any) -> None:
        with self._cv:
            if len(self._buf) == self._buf.maxlen:
                self._offset0 += 1
            self._buf.append(value)
            self._cv.notify_all()
        for cb in list(self._subscribers):
            cb(value)

    def subscribe(self, callback: Callable[[Any], None]) -> None:
        with self._cv:
            self._subscribers = self._subscribers + [callback]

    def end_offset(self) -> int:
        with self._cv:
            return self._offset0 + len(self._buf)

    def read(self, offset: int, timeout: Optional[float] = None) -> Any:
        """Blocking read of the message at `offset`; None on timeout."""
        with self._cv:
            while self._offset0 + len(self._buf) <= offset:
                if not self._cv.wait(timeout=timeout):
                    return None
            return self._buf[offset - self._offset0]


class MessageBus:
    """Named topics with the reference topic set pre-created."""

    def __init__(self):
        self.topics: Dict[str, Topic] = {t: Topic(t) for t in TOPICS}

    def topic(self, name: str) -> Topic:
        if name not in self.topics:
            self.topics[name] = Topic(name)
        return self.topics[name]

    def publish(self, topic: str, value: Any) -> None:
        self.topic(topic).publish(value)

    def consume(self, topic: str, from_end: bool = True,
                timeout: Optional[float] = None) -> Iterator[Any]:
        """Iterator over messages, starting at the end like the reference
        consumer's seek_to_end (predict.py:30). The end offset is captured
        HERE (consumer creation), not lazily at the first next() — a
        generator body would only run on first iteration and would skip
        messages published in between."""
        t = self.topic(topic)
        offset = t.end_offset() if from_end else 0

        def _iter():
            nonlocal offset
            while True:
                msg = t.read(offset, timeout=timeout)
                if msg is None:
                    return
                offset += 1
                yield msg

        return _iter()
