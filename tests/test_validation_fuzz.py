"""Property test: the validating webhook either accepts or raises
ValidationError — never an uncontrolled TypeError/AttributeError —
for ARBITRARY spec shapes (clients send anything over the API)."""

import pytest

hypothesis = pytest.importorskip("hypothesis")
from hypothesis import given, settings, strategies as st  # noqa: E402

from datatunerx_amd.api.types import (Dataset, Finetune,  # noqa: E402
                                      FinetuneExperiment, FinetuneJob,
                                      Hyperparameter, LLM, Scoring)
from datatunerx_amd.api.validation import (ValidationError,  # noqa: E402
                                           default_, validate_)

json_scalars = st.one_of(st.none(), st.booleans(),
                         st.integers(-10**6, 10**6),
                         st.floats(allow_nan=False, allow_infinity=False),
                         st.text(max_size=20))
json_values = st.recursive(
    json_scalars,
    lambda children: st.one_of(
        st.lists(children, max_size=4),
        st.dictionaries(st.text(max_size=12), children, max_size=4)),
    max_leaves=12)
specs = st.dictionaries(
    st.sampled_from(["fineTune", "finetuneSpec", "llm", "dataset",
                     "hyperparameter", "parameters", "node", "pending",
                     "finetuneJobs", "datasetMetadata", "datasetInfo",
                     "subsets", "features", "serveConfig",
                     "tensorParallel", "scoringPluginConfig", "plugin",
                     "stage", "epochs", "batchSize", "int4", "int8"]) |
    st.text(max_size=12),
    json_values, max_size=6)


@settings(max_examples=150, deadline=None)
@given(kind=st.sampled_from([FinetuneJob, FinetuneExperiment, Finetune,
                             Hyperparameter, Dataset, LLM, Scoring]),
       name=st.text(max_size=20), spec=specs)
def test_validate_never_crashes(kind, name, spec):
    obj = kind(name=name, spec=spec)
    try:
        default_(obj)
        validate_(obj)
    except ValidationError:
        pass            # the controlled rejection path
    except (AttributeError, TypeError, KeyError) as e:
        pytest.fail(f"uncontrolled {type(e).__name__} for "
                    f"{kind.__name__} spec={spec!r}: {e}")


@settings(max_examples=120, deadline=None)
@given(tmpl=st.sampled_from(sorted(
           __import__("datatunerx_amd.data.templates",
                      fromlist=["TEMPLATES"]).TEMPLATES)),
       query=st.text(max_size=60), resp=st.text(max_size=40),
       system=st.text(max_size=30),
       history=st.lists(st.tuples(st.text(max_size=20),
                                  st.text(max_size=20)), max_size=3),
       cutoff=st.integers(4, 200))
def test_preprocess_never_crashes_and_masks_consistently(
        tmpl, query, resp, system, history, cutoff):
    """Property: template encode + masking hold for ARBITRARY text —
    ids/labels same length, bounded by cutoff, every non-ignored label
    equals its input id."""
    from datatunerx_amd.data.dataset import (IGNORE_INDEX, ByteTokenizer,
                                             preprocess_supervised_example)
    ids, labels = preprocess_supervised_example(
        ByteTokenizer(), tmpl, query, resp, history=history,
        system=system, cutoff_len=cutoff)
    assert len(ids) == len(labels) <= cutoff
    for i, l in zip(ids, labels):
        assert l == IGNORE_INDEX or l == i


@settings(max_examples=100, deadline=None)
@given(batch=st.lists(
           st.lists(st.integers(0, 500), min_size=1, max_size=40),
           min_size=1, max_size=6),
       mult=st.integers(1, 8))
def test_collate_invariants(batch, mult):
    from datatunerx_amd.data.dataset import IGNORE_INDEX, collate
    exs = [{"input_ids": ids, "labels": list(ids)} for ids in batch]
    out = collate(exs, pad_token_id=0, pad_to_multiple_of=mult)
    B, S = out["input_ids"].shape
    assert B == len(batch) and S % mult == 0
    assert S >= max(len(b) for b in batch)
    for i, ids in enumerate(batch):
        n = len(ids)
        assert out["input_ids"][i, :n].tolist() == ids
        assert (out["input_ids"][i, n:] == 0).all()
        assert (out["labels"][i, n:] == IGNORE_INDEX).all()


@settings(max_examples=100, deadline=None)
@given(pred=st.lists(st.integers(0, 50), max_size=30),
       ref=st.lists(st.integers(0, 50), max_size=30))
def test_gen_metrics_bounded(pred, ref):
    from datatunerx_amd.train.gen_metrics import bleu, rouge_l
    for m in (rouge_l(pred, ref), bleu(pred, ref)):
        assert 0.0 <= m <= 1.0
    if pred and pred == ref:
        assert rouge_l(pred, ref) == 1.0
