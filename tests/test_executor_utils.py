"""Executor hyperparameter + data-path registries (reference parity:
executors/accelerate/.../utils.py get_loss_fn/get_scheduler and
dataset.py IterableStreamDataSet/dataset_wrapper)."""

import math

import pytest
import torch

from hypha_amd.data.stream import SliceStreamDataset, build_preprocessor, infinite
from hypha_amd.data.synthetic import write_slice_files
from hypha_amd.parallel import InnerOptConfig, lr_at
from hypha_amd.runtime.losses import apply_wire_schedule, get_loss_fn


def test_loss_registry_matches_reference_set():
    # the five wire losses (messages lib.rs:662-670)
    assert isinstance(get_loss_fn("l1"), torch.nn.L1Loss)
    assert isinstance(get_loss_fn("mse"), torch.nn.MSELoss)
    assert isinstance(get_loss_fn("cross-entropy"), torch.nn.CrossEntropyLoss)
    assert isinstance(get_loss_fn("bce-with-logits"), torch.nn.BCEWithLogitsLoss)
    assert isinstance(get_loss_fn("kl-div"), torch.nn.KLDivLoss)
    with pytest.raises(ValueError, match="not supported"):
        get_loss_fn("hinge")


def test_wire_schedule_mapping():
    cfg = InnerOptConfig(lr=1.0)
    apply_wire_schedule(cfg, None)
    assert cfg.schedule == "constant"
    assert lr_at(cfg, cfg.warmup_steps + 5) == 1.0

    apply_wire_schedule(cfg, {"type": "cosine-with-warmup", "warmup_steps": 4,
                              "training_steps": 100})
    assert cfg.schedule == "cosine" and cfg.warmup_steps == 4 and cfg.total_steps == 100
    assert lr_at(cfg, 0) == pytest.approx(0.25)  # warmup ramp
    # cosine midpoint: lr = min + (1-min)*0.5
    mid = lr_at(cfg, 4 + 48)
    assert mid == pytest.approx(cfg.min_lr_frac + (1 - cfg.min_lr_frac) * 0.5, abs=0.02)

    apply_wire_schedule(cfg, {"type": "linear-with-warmup", "warmup_steps": 0,
                              "training_steps": 10})
    assert lr_at(cfg, 10) == pytest.approx(cfg.min_lr_frac)

    apply_wire_schedule(cfg, {"type": "wsd", "warmup_steps": 10, "decay_step": 100})
    # decay begins at ~decay_step: stable just before, decaying just after
    assert lr_at(cfg, 95) == pytest.approx(1.0)
    assert lr_at(cfg, 105) < 1.0

    with pytest.raises(ValueError, match="not supported"):
        apply_wire_schedule(cfg, {"type": "step"})


def test_slice_stream_dataset(tmp_path):
    paths = write_slice_files(str(tmp_path), "ds", num_slices=2,
                              samples_per_slice=3, vocab_size=50, seq_len=8)
    ds = SliceStreamDataset(iter(paths), model_inputs=["input_ids"])
    samples = list(ds)
    assert len(samples) == 6
    assert samples[0]["input_ids"].shape == (8,)

    # preprocessor over processor_inputs replaces those keys with its output
    def upper_bound(**kw):
        return {"input_ids": kw["input_ids"].clamp(max=10)}

    ds2 = SliceStreamDataset(iter(paths), model_inputs=["input_ids"],
                             processor_inputs=["input_ids"], preprocessor=upper_bound)
    assert all(s["input_ids"].max() <= 10 for s in ds2)

    # endless-epoch wrapper (dataset.py:37-41)
    it = infinite(SliceStreamDataset(iter(paths), model_inputs=["input_ids"]))
    # the wrapped iterator re-enters __iter__, but the path iterator is spent;
    # pass a re-iterable list for true epochs
    it = infinite(SliceStreamDataset(paths, model_inputs=["input_ids"]))
    first_epoch = [next(it) for _ in range(6)]
    second_epoch = [next(it) for _ in range(6)]
    assert torch.equal(first_epoch[0]["input_ids"], second_epoch[0]["input_ids"])


def test_preprocessor_registry_errors():
    with pytest.raises(ValueError, match="not supported"):
        build_preprocessor("audio")
    with pytest.raises(FileNotFoundError, match="no network"):
        build_preprocessor("tokenizer", "/nonexistent/tok")
    with pytest.raises(FileNotFoundError, match="no network"):
        build_preprocessor("image")  # media types need a local artifact too


def test_tokenizer_preprocessor_from_local_artifact(tmp_path):
    """The wire PreprocessorType 'tokenizer' resolved against a LOCAL
    artifact (offline parity for the reference's HF preprocessor fetch)."""
    tokenizers = pytest.importorskip("tokenizers")
    transformers = pytest.importorskip("transformers")
    from tokenizers.pre_tokenizers import Whitespace

    tok = tokenizers.Tokenizer(
        tokenizers.models.WordLevel({"hello": 0, "world": 1, "[UNK]": 2},
                                    unk_token="[UNK]"))
    tok.pre_tokenizer = Whitespace()
    fast = transformers.PreTrainedTokenizerFast(tokenizer_object=tok,
                                                pad_token="[UNK]")
    art = tmp_path / "tok"
    fast.save_pretrained(str(art))

    run = build_preprocessor("tokenizer", str(art))
    out = run(text=["hello world", "world"])
    assert out["input_ids"].shape[0] == 2
    assert out["input_ids"][0].tolist()[:2] == [0, 1]


def test_lr_schedules_vs_transformers_oracle():
    """Shape parity against the SAME schedule implementations the reference
    executor uses (utils.py:90-106 -> transformers get_*_schedule_with_warmup):
    warmup ramp and decay trajectory agree within tolerance (our cosine/linear
    floors at min_lr_frac, the oracle at 0 — compare above the floor)."""
    import torch as t
    from transformers import (get_cosine_schedule_with_warmup,
                              get_linear_schedule_with_warmup)

    warmup, total = 10, 100

    def oracle_lrs(make):
        opt = t.optim.SGD([t.nn.Parameter(t.zeros(1))], lr=1.0)
        sched = make(opt)
        out = []
        for _ in range(total):
            out.append(opt.param_groups[0]["lr"])
            opt.step()
            sched.step()
        return out

    cos = oracle_lrs(lambda o: get_cosine_schedule_with_warmup(o, warmup, total))
    lin = oracle_lrs(lambda o: get_linear_schedule_with_warmup(o, warmup, total))

    cfg = InnerOptConfig(lr=1.0, min_lr_frac=0.0)
    apply_wire_schedule(cfg, {"type": "cosine-with-warmup", "warmup_steps": warmup,
                              "training_steps": total})
    ours_cos = [lr_at(cfg, i) for i in range(total)]
    apply_wire_schedule(cfg, {"type": "linear-with-warmup", "warmup_steps": warmup,
                              "training_steps": total})
    ours_lin = [lr_at(cfg, i) for i in range(total)]

    # warmup: both ramp 0->1 linearly over `warmup` steps (off-by-one
    # conventions differ by <= 1/warmup)
    for i in range(warmup):
        assert abs(ours_cos[i] - cos[i]) <= 1.0 / warmup + 1e-9
    # decay: same trajectory within a coarse tolerance (conventions differ
    # in the (step - warmup)/(total - warmup) vs step/total denominators)
    for i in range(warmup + 1, total, 7):
        assert abs(ours_cos[i] - cos[i]) < 0.08, (i, ours_cos[i], cos[i])
        assert abs(ours_lin[i] - lin[i]) < 0.08, (i, ours_lin[i], lin[i])
    # endpoints agree exactly: peak 1.0 after warmup, ~0 at the end
    assert ours_cos[-1] < 0.01 and cos[-1] < 0.01
    assert ours_lin[-1] < 0.02 and lin[-1] < 0.02


def test_feature_preprocessor_from_local_artifact(tmp_path):
    """PreprocessorType::Feature resolved against a fetched local artifact
    (reference utils.py:45-46 AutoFeatureExtractor mapping) — offline."""
    import json

    import torch

    from hypha_amd.data.stream import build_preprocessor

    art = tmp_path / "fe"
    art.mkdir()
    (art / "preprocessor_config.json").write_text(json.dumps({
        "feature_extractor_type": "Wav2Vec2FeatureExtractor",
        "feature_size": 1, "sampling_rate": 16000, "padding_value": 0.0,
        "return_attention_mask": False, "do_normalize": True}))
    pre = build_preprocessor("feature", str(art))
    out = pre(audio=torch.randn(2, 400))
    assert "input_values" in out and out["input_values"].shape == (2, 400)
    # normalized output
    assert abs(float(out["input_values"][0].mean())) < 0.1


def test_image_preprocessor_from_local_artifact(tmp_path):
    import json

    import torch

    from hypha_amd.data.stream import build_preprocessor

    art = tmp_path / "im"
    art.mkdir()
    (art / "preprocessor_config.json").write_text(json.dumps({
        "image_processor_type": "ViTImageProcessor", "do_resize": True,
        "size": {"height": 32, "width": 32}, "do_normalize": True,
        "image_mean": [0.5, 0.5, 0.5], "image_std": [0.5, 0.5, 0.5],
        "do_rescale": True}))
    pre = build_preprocessor("image", str(art))
    imgs = torch.randint(0, 255, (2, 3, 48, 48), dtype=torch.uint8)
    out = pre(images=imgs)
    assert out["pixel_values"].shape == (2, 3, 32, 32)


def test_preprocessor_missing_artifact_raises():
    import pytest as _pytest

    from hypha_amd.data.stream import build_preprocessor

    with _pytest.raises(FileNotFoundError):
        build_preprocessor("feature", "/nonexistent/path")
