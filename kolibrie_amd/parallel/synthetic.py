"""Synthetic employee-shaped RDF generator ("synthetic employee
triples"; ref kolibrie/examples/synthetic_data/gen_data.rs — employees with
foaf:name / ds:position / ds:annual_salary / foaf:workplaceHomepage /
ds:worksFor, plus departments with ds:locatedIn).

Generates dictionary-encoded int32 columns directly (no string round trip:
entity/literal IDs are allocated in dense blocks above the interned
vocabulary), optionally hash-partitioned by subject for multi-GPU ranks.
"""
from __future__ import annotations

from dataclasses import dataclass
from typing import Dict, Tuple

import numpy as np
import torch

FOAF = "http://xmlns.com/foaf/0.1/"
DS = "https://data.cityofchicago.org/resource/xzkq-xp2w/"

PREDICATES = {
    "name": FOAF + "name",
    "homepage": FOAF + "workplaceHomepage",
    "salary": DS + "annual_salary",
    "position": DS + "position",
    "email": DS + "email",
    "age": DS + "age",
    "worksFor": DS + "worksFor",
    "locatedIn": DS + "locatedIn",
    "label": "http://www.w3.org/2000/01/rdf-schema#label",
}

TRIPLES_PER_EMP = 7
TRIPLES_PER_DEPT = 2
DEPT_RATIO = 100  # employees per department
N_POSITIONS = 3
N_CITIES = 1000


@dataclass
class EmployeeDataset:
    pred_ids: Dict[str, int]
    position_ids: Tuple[int, ...]
    n_employees: int
    n_departments: int
    emp_base: int
    dept_base: int
    name_base: int
    salary_base: int
    city_base: int
    n_salary_values: int


def plan_dataset(db, total_triples: int) -> EmployeeDataset:
    """Intern predicates and allocate dense ID blocks for entities."""
    n_emp = max(1, total_triples // (TRIPLES_PER_EMP + TRIPLES_PER_DEPT / DEPT_RATIO))
    n_emp = int(n_emp)
    n_dept = max(1, n_emp // DEPT_RATIO)
    pred_ids = {k: db.dictionary.encode(v) for k, v in PREDICATES.items()}
    position_ids = tuple(
        db.dictionary.encode(n) for n in ("Manager", "Developer", "Salesperson"))
    base = len(db.dictionary) + 64
    emp_base = base
    dept_base = emp_base + n_emp
    name_base = dept_base + n_dept
    salary_base = name_base + n_emp
    n_salary = 120_000
    city_base = salary_base + n_salary
    return EmployeeDataset(
        pred_ids=pred_ids, position_ids=position_ids,
        n_employees=n_emp, n_departments=n_dept,
        emp_base=emp_base, dept_base=dept_base, name_base=name_base,
        salary_base=salary_base, city_base=city_base, n_salary_values=n_salary,
    )


def generate_partition(ds: EmployeeDataset, rank: int, world: int, seed: int,
                       device, replicate_dept: bool = False
                       ) -> Tuple[torch.Tensor, torch.Tensor, torch.Tensor]:
    """Generate this rank's subject-partitioned shard as (s,p,o) int32 cols.

    Partition function: subject_id % world (a valid hash partition over the
    dense synthetic ID blocks) — employees and departments both shard by
    their own subject id, exactly what a distributed loader would do.
    `replicate_dept=True` keeps the small department relation on EVERY rank
    (the broadcast-table layout a distributed planner chooses for tiny
    build sides: join traffic becomes zero).
    """
    dev = torch.device(device)
    gen = torch.Generator(device=dev)
    gen.manual_seed(seed + 7919 * rank)
    emp_all = torch.arange(ds.n_employees, dtype=torch.int64, device=dev)
    emp = emp_all[(ds.emp_base + emp_all) % world == rank]
    ne = emp.numel()
    p = ds.pred_ids

    s_parts, p_parts, o_parts = [], [], []

    def add(pred_key: str, subjects: torch.Tensor, objects: torch.Tensor):
        s_parts.append(subjects.to(torch.int32))
        p_parts.append(torch.full((subjects.numel(),), p[pred_key],
                                  dtype=torch.int32, device=dev))
        o_parts.append(objects.to(torch.int32))

    def rnd(hi: int, n: int) -> torch.Tensor:
        return torch.randint(0, hi, (n,), generator=gen, dtype=torch.int64,
                             device=dev)

    emp_ids = ds.emp_base + emp
    add("name", emp_ids, ds.name_base + emp)
    add("homepage", emp_ids, ds.name_base + emp)  # homepage shares name id pool
    add("salary", emp_ids, ds.salary_base + rnd(ds.n_salary_values, ne))
    pos_tbl = torch.tensor(ds.position_ids, dtype=torch.int64, device=dev)
    add("position", emp_ids, pos_tbl[rnd(N_POSITIONS, ne)])
    add("email", emp_ids, ds.name_base + emp)
    add("age", emp_ids, ds.salary_base + rnd(50, ne))
    dept_of = ds.dept_base + (emp % ds.n_departments)
    add("worksFor", emp_ids, dept_of)

    dept = torch.arange(ds.n_departments, dtype=torch.int64, device=dev)
    if not replicate_dept:
        dept = dept[(ds.dept_base + dept) % world == rank]
    dept_ids = ds.dept_base + dept
    add("locatedIn", dept_ids, ds.city_base + (dept % N_CITIES))
    add("label", dept_ids, ds.name_base + (dept % max(1, ds.n_employees)))

    return torch.cat(s_parts), torch.cat(p_parts), torch.cat(o_parts)


FLAGSHIP_QUERY = f"""
PREFIX ds: <{DS}>
SELECT (COUNT(*) AS ?c) WHERE {{
    ?e ds:worksFor ?d .
    ?e ds:annual_salary ?sal .
    ?d ds:locatedIn ?city .
}}
"""
