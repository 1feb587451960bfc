"""Event counters and the headline messages-saved metric.

Counting rule: the always-communicate ring sends
``2 * num_tensors`` messages per rank per pass; EventGraD counts
``num_events += 2`` per fired tensor (event.cpp:344). Then

    saved% = 1 - total_events / (2 * num_tensors * total_passes * world)

The first ``initial_comm_passes`` (30) passes always fire (event.cpp:343).
"""

from __future__ import annotations

from dataclasses import dataclass, field


@dataclass
class RunMetrics:
    world: int = 1
    num_tensors: int = 0
    total_passes: int = 0           # per-rank pass count
    num_events_total: int = 0       # allreduced across ranks
    train_time_s: float = 0.0
    samples_seen: int = 0           # per-rank
    epoch_train_acc: list = field(default_factory=list)
    test_accuracy: float | None = None
    test_loss: float | None = None
    final_train_loss: float | None = None

    @property
    def messages_possible(self) -> int:
        return 2 * self.num_tensors * self.total_passes * self.world

    @property
    def messages_saved_pct(self) -> float:
        possible = self.messages_possible
        if possible == 0:
            return 0.0
        return 100.0 * (1.0 - self.num_events_total / possible)

    @property
    def samples_per_sec(self) -> float:
        if self.train_time_s <= 0:
            return 0.0
        return self.world * self.samples_seen / self.train_time_s

    def summary(self) -> dict:
        return {
            "world": self.world,
            "total_passes": self.total_passes,
            "num_events_total": self.num_events_total,
            "messages_possible": self.messages_possible,
            "messages_saved_pct": round(self.messages_saved_pct, 3),
            "train_time_s": round(self.train_time_s, 3),
            "samples_per_sec": round(self.samples_per_sec, 2),
            "epoch_train_acc": self.epoch_train_acc,
            "final_train_loss": self.final_train_loss,
            "test_accuracy": self.test_accuracy,
            "test_loss": self.test_loss,
        }
