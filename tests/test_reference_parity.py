# -*- coding: utf-8 -*-
"""Automated parity proof against the reference's public config API.

Loads the reference ``stoke/configs.py`` by file path (its only imports are
attrs/torch/typing, so no DeepSpeed/fairscale/Horovod install is needed) and
compares every attrs class: presence, field names, field order, and default
values.  Skipped wherever the reference checkout is not present.

Known intentional deviations are listed explicitly below.
"""

import importlib.util
import os

import attr
import pytest

REF = "/root/reference/stoke/configs.py"

# field -> (class, reason) for deliberate default differences
INTENTIONAL = {
    ("DDPConfig", "bucket_cap_mb"):
        "64 MB vs 25: tuned for the 7-link xGMI mesh (SURVEY.md 5.8)",
    ("BackendOptions", "mpi"):
        "reference enum value ' mpi' has a stray leading space (latent bug)",
}


def _load_reference():
    if not os.path.exists(REF):
        pytest.skip("reference checkout not present")
    spec = importlib.util.spec_from_file_location("ref_configs", REF)
    mod = importlib.util.module_from_spec(spec)
    spec.loader.exec_module(mod)
    return mod


def test_config_classes_and_defaults_match_reference():
    ref = _load_reference()
    import stoke.configs as ours

    ref_classes = {
        name: cls
        for name, cls in vars(ref).items()
        if attr.has(cls) if isinstance(cls, type)
    }
    assert ref_classes, "reference attrs classes not found"
    mismatches = []
    for name, rcls in ref_classes.items():
        ocls = getattr(ours, name, None)
        if ocls is None:
            mismatches.append(f"missing class {name}")
            continue
        rf = {f.name: f for f in attr.fields(rcls)}
        of = {f.name: f for f in attr.fields(ocls)}
        if set(rf) != set(of):
            mismatches.append(
                f"{name}: fields differ missing={set(rf)-set(of)} "
                f"extra={set(of)-set(rf)}"
            )
            continue
        for fname, fld in rf.items():
            if (name, fname) in INTENTIONAL:
                continue
            rdef, odef = fld.default, of[fname].default
            if type(rdef).__name__ == "_Nothing":  # required field
                continue
            if _norm(rdef) != _norm(odef):
                mismatches.append(
                    f"{name}.{fname}: default {odef!r} != reference {rdef!r}"
                )
    assert not mismatches, "\n".join(mismatches)


def _norm(v):
    """Normalize a default for comparison: recurse into attrs instances,
    unwrap enums, strip strings (the reference's ' mpi' bug)."""
    if attr.has(type(v)):
        return {f.name: _norm(getattr(v, f.name)) for f in attr.fields(type(v))}
    v = getattr(v, "value", v)
    if isinstance(v, str):
        return v.strip()
    return v


def test_enums_match_reference():
    ref = _load_reference()
    import stoke.configs as ours
    from enum import Enum

    for ename in ("HorovodOps", "OffloadDevice", "BackendOptions"):
        rcls = getattr(ref, ename)
        ocls = getattr(ours, ename)
        assert issubclass(ocls, Enum)
        rnames = [m.name for m in rcls]
        onames = [m.name for m in ocls]
        assert rnames == onames, f"{ename}: {onames} != {rnames}"


def test_sampler_lengths_match_reference():
    """Per-replica sample counts equal the reference's
    BucketedDistributedSampler across a parameter grid (including the
    bucket-overlap residual batches)."""
    import sys
    import types

    if not os.path.exists("/root/reference/stoke/data.py"):
        pytest.skip("reference checkout not present")
    # the reference imports horovod unconditionally; stub it
    if "horovod" not in sys.modules:
        h = types.ModuleType("horovod")
        ht = types.ModuleType("horovod.torch")
        h.torch = ht
        sys.modules["horovod"] = h
        sys.modules["horovod.torch"] = ht
    spec = importlib.util.spec_from_file_location(
        "ref_data", "/root/reference/stoke/data.py"
    )
    mod = importlib.util.module_from_spec(spec)
    spec.loader.exec_module(mod)
    from stoke.data import BucketedDistributedSampler as Ours

    ds = list(range(1000))
    si = list(range(1000))
    for buckets, batch, reps, dl, overlap in [
        (2, 4, 2, False, False),
        (2, 4, 2, True, False),
        (4, 8, 4, True, False),
        (2, 8, 2, False, False),
        (4, 8, 4, True, True),
    ]:
        kw = dict(buckets=buckets, batch_size=batch, sorted_idx=si,
                  backend=None, allow_bucket_overlap=overlap,
                  num_replicas=reps, rank=0, drop_last=dl)
        r = mod.BucketedDistributedSampler(ds, **kw)
        o = Ours(ds, **kw)
        assert len(r) == len(o), (buckets, batch, reps, dl, overlap)
        assert len(list(iter(r))) == len(list(iter(o)))
