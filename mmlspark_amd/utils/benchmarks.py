"""Accuracy-benchmark regression harness (Benchmarks trait parity,
core/.../core/test/benchmarks/Benchmarks.scala:36-85): tests compute metrics,
append them to new_benchmarks/, and compare against a committed CSV with
per-entry precision and direction."""
from __future__ import annotations

import os
from dataclasses import dataclass
from typing import List

import pandas as pd


@dataclass
class BenchmarkEntry:
    name: str
    value: float
    precision: float = 0.0   # tolerance; 0 = exact
    higher_is_better: bool = True


class BenchmarkRunner:
    def __init__(self, suite: str, resource_dir: str,
                 new_dir: str = "new_benchmarks"):
        self.suite = suite
        self.resource = os.path.join(resource_dir, f"benchmarks_{suite}.csv")
        self.new_path = os.path.join(new_dir, f"benchmarks_{suite}.csv")
        self.entries: List[BenchmarkEntry] = []

    def add(self, name: str, value: float, precision: float = 0.0,
            higher_is_better: bool = True):
        self.entries.append(BenchmarkEntry(name, value, precision,
                                           higher_is_better))

    def write_new(self):
        os.makedirs(os.path.dirname(self.new_path) or ".", exist_ok=True)
        pd.DataFrame([e.__dict__ for e in self.entries]).to_csv(
            self.new_path, index=False)

    def compare(self) -> List[str]:
        """Return list of violations vs the committed CSV (empty = pass).
        A measured value BETTER than committed (within direction) passes."""
        self.write_new()
        if not os.path.exists(self.resource):
            raise FileNotFoundError(
                f"committed benchmark file missing: {self.resource}; "
                f"copy {self.new_path} there to establish the baseline")
        ref = pd.read_csv(self.resource).set_index("name")
        problems = []
        for e in self.entries:
            if e.name not in ref.index:
                problems.append(f"{e.name}: not in committed baseline")
                continue
            row = ref.loc[e.name]
            expected = float(row["value"])
            tol = float(row.get("precision", e.precision))
            if e.higher_is_better:
                if e.value < expected - tol:
                    problems.append(
                        f"{e.name}: {e.value:.4f} < {expected:.4f} - {tol}")
            else:
                if e.value > expected + tol:
                    problems.append(
                        f"{e.name}: {e.value:.4f} > {expected:.4f} + {tol}")
        return problems
