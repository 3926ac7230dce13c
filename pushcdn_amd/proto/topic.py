"""Topic space: a topic is one u8 on the wire; applications prune to their own
enum (reference ``cdn-proto/src/def.rs:23-50``)."""

from __future__ import annotations

from typing import Iterable, List, Sequence

from .errors import TopicError


class TopicSpace:
    """A set of valid topic ids with the reference's prune semantics
    (def.rs:31-50): dedupe, drop unknown ids, error if nothing valid remains."""

    def __init__(self, valid: Iterable[int]) -> None:
        self.valid = frozenset(int(t) & 0xFF for t in valid)

    def prune(self, topics: Sequence[int]) -> List[int]:
        seen = set()
        out: List[int] = []
        for t in topics:
            t = int(t) & 0xFF
            if t in self.valid and t not in seen:
                seen.add(t)
                out.append(t)
        if not out:
            raise TopicError("no valid topics after pruning")
        return out


# The reference's test topic space (def.rs:23-28): Global=0, DA=1.
class TestTopic:
    GLOBAL = 0
    DA = 1


TEST_TOPIC_SPACE = TopicSpace([TestTopic.GLOBAL, TestTopic.DA])

# A full 256-topic space for benchmarks / production embedders.
ALL_TOPICS = TopicSpace(range(256))
