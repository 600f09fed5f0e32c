"""Hierarchical reasoning (experimental).

Ref parity: datalog/src/reasoning_experimental.rs (307 LoC) —
ReasoningLevel {Base, Deductive, Abductive, MetaReasoning} (:18), one
Reasoner per level, HierarchicalRule with priority + level dependencies,
hierarchical_inference: per-level semi-naive then cross-level rules
(:86-160).
"""
from __future__ import annotations

from dataclasses import dataclass, field
from typing import Dict, List

from .reasoner import Reasoner
from .rule import Rule


class ReasoningLevel:
    BASE = 0
    DEDUCTIVE = 1
    ABDUCTIVE = 2
    META_REASONING = 3

    ALL = (BASE, DEDUCTIVE, ABDUCTIVE, META_REASONING)
    NAMES = {0: "Base", 1: "Deductive", 2: "Abductive", 3: "MetaReasoning"}


@dataclass
class HierarchicalRule:
    rule: Rule
    level: int
    priority: int = 0
    depends_on: List[int] = field(default_factory=list)


class ReasoningHierarchy:
    """One reasoner per level; facts flow upward through level inference
    then cross-level rules run on the combined store."""

    def __init__(self, device: str = "cpu"):
        from ..storage.dictionary import Dictionary
        self.dictionary = Dictionary()
        self.levels: Dict[int, Reasoner] = {
            lv: Reasoner(device=device, dictionary=self.dictionary)
            for lv in ReasoningLevel.ALL
        }
        self.rules: List[HierarchicalRule] = []

    def add_fact(self, level: int, s: str, p: str, o: str):
        self.levels[level].add_abox_triple(s, p, o)

    def add_rule(self, hr: HierarchicalRule):
        self.rules.append(hr)

    def hierarchical_inference(self) -> int:
        """Per-level semi-naive in level order (respecting priorities),
        then cross-level rules over facts promoted from dependency levels
        (ref :86-160).  Returns total facts derived."""
        total = 0
        for lv in ReasoningLevel.ALL:
            r = self.levels[lv]
            level_rules = sorted(
                (hr for hr in self.rules if hr.level == lv and not hr.depends_on),
                key=lambda hr: -hr.priority)
            r.rules = [hr.rule for hr in level_rules]
            r.rule_index = type(r.rule_index)()
            for rule in r.rules:
                r.rule_index.add_rule(rule)
            total += r.infer_new_facts_semi_naive()
        # cross-level: promote dependency facts, run dependent rules
        for hr in sorted((h for h in self.rules if h.depends_on),
                         key=lambda h: -h.priority):
            target = self.levels[hr.level]
            for dep in hr.depends_on:
                for (s, p, o) in self.levels[dep].all_fact_tuples():
                    target.add_fact_ids(s, p, o)
            target.rules = [hr.rule]
            target.rule_index = type(target.rule_index)()
            target.rule_index.add_rule(hr.rule)
            total += target.infer_new_facts_semi_naive()
        return total

    def query_level(self, level: int, s=None, p=None, o=None):
        return self.levels[level].query_abox(s, p, o)
