"""Staged disagg profile-handler state machine, table-driven after the
reference's scenario matrix (disagg_profile_handler_test.go:364-892:
decode-not-run/decode-failed/encode-per-modality/fall-throughs/decision
labels)."""
import pytest

from llm_d_inference_scheduler_amd.datalayer.datastore import make_endpoint
from llm_d_inference_scheduler_amd.plugins.profile_handlers import (
    ENCODER_HEADER, PREFILLER_HEADER, AlwaysDisaggMultimodalDecider,
    AlwaysDisaggPDDecider, DisaggProfileHandler, PrefixBasedPDDecider)
from llm_d_inference_scheduler_amd.scheduling.types import (
    LLMRequest, MultiModalItem, ProfileRunResult, SchedulingContext,
    SchedulingResult)

PROFILES = {"decode": object(), "prefill": object(), "encode": object()}


def ctx_with(prompt_tokens=64, mm=None):
    req = LLMRequest(request_id="r", model="m", prompt="x" * 16,
                     prompt_tokens=list(range(prompt_tokens)), max_tokens=4,
                     mm_items=[MultiModalItem(k, "u") for k in (mm or [])])
    return SchedulingContext(request=req)


def res(name, picked=True):
    r = ProfileRunResult(profile_name=name)
    if picked:
        r.picks = [make_endpoint(f"{name}-ep", 0)]
        r.scores = {f"{name}-ep": 1.0}
    return r


def handler(pd=True, enc=False):
    return DisaggProfileHandler(
        "h", pdDecider=AlwaysDisaggPDDecider("pd") if pd else None,
        encodeDecider=AlwaysDisaggMultimodalDecider("mm") if enc else None)


class TestStagedPicks:
    def test_decode_not_run_runs_decode(self):
        assert handler().pick_profiles(ctx_with(), PROFILES, {}) == ["decode"]

    def test_decode_failed_done(self):
        results = {"decode": res("decode", picked=False)}
        assert handler().pick_profiles(ctx_with(), PROFILES, results) == []

    def test_pd_decider_fires_prefill(self):
        results = {"decode": res("decode")}
        assert handler().pick_profiles(ctx_with(), PROFILES, results) == \
            ["prefill"]

    @pytest.mark.parametrize("mm", [["image_url"], ["video_url"],
                                    ["input_audio"]])
    def test_multimodal_runs_encode(self, mm):
        h = handler(pd=True, enc=True)
        results = {"decode": res("decode")}
        assert h.pick_profiles(ctx_with(mm=mm), PROFILES, results) == \
            ["encode", "prefill"]

    def test_text_only_skips_encode(self):
        h = handler(pd=True, enc=True)
        results = {"decode": res("decode")}
        assert h.pick_profiles(ctx_with(), PROFILES, results) == ["prefill"]

    def test_all_done_returns_empty(self):
        h = handler(pd=True, enc=True)
        results = {"decode": res("decode"), "prefill": res("prefill"),
                   "encode": res("encode")}
        assert h.pick_profiles(ctx_with(mm=["image_url"]), PROFILES,
                               results) == []

    def test_short_uncached_suffix_skips_prefill(self):
        h = DisaggProfileHandler(
            "h", pdDecider=PrefixBasedPDDecider("p", nonCachedTokens=512))
        results = {"decode": res("decode")}
        assert h.pick_profiles(ctx_with(prompt_tokens=64), PROFILES,
                               results) == []

    def test_long_uncached_prompt_triggers_prefill(self):
        h = DisaggProfileHandler(
            "h", pdDecider=PrefixBasedPDDecider("p", nonCachedTokens=512))
        results = {"decode": res("decode")}
        assert h.pick_profiles(ctx_with(prompt_tokens=1024), PROFILES,
                               results) == ["prefill"]


class TestProcessAndHeaders:
    @pytest.mark.parametrize("have,decision", [
        (["decode"], "decode_only"),
        (["decode", "prefill"], "pd"),
        (["decode", "prefill", "encode"], "epd"),
        (["decode", "encode"], "e_pd"),
    ])
    def test_decision_labels(self, have, decision):
        h = handler(pd=True, enc=True)
        ctx = ctx_with(mm=["image_url"])
        results = {n: res(n) for n in have}
        primary = h.process_results(ctx, results)
        assert primary == "decode"
        assert ctx.state["disagg_decision"] == decision

    def test_rejected_encode_omitted_from_headers(self):
        """encode nil (rejected) -> omitted (reference :782)."""
        h = handler(pd=True, enc=True)
        ctx = ctx_with(mm=["image_url"])
        results = {"decode": res("decode"), "prefill": res("prefill"),
                   "encode": res("encode", picked=False)}
        h.process_results(ctx, results)
        assert ctx.state["disagg_decision"] == "pd"
        sr = SchedulingResult(profile_results=results,
                              primary_profile="decode")
        h.pre_request(ctx, sr, results["decode"].target)
        assert PREFILLER_HEADER in ctx.request.headers
        assert ENCODER_HEADER not in ctx.request.headers

    def test_stage_headers_published(self):
        h = handler(pd=True, enc=True)
        ctx = ctx_with(mm=["image_url"])
        results = {"decode": res("decode"), "prefill": res("prefill"),
                   "encode": res("encode")}
        sr = SchedulingResult(profile_results=results,
                              primary_profile="decode")
        h.pre_request(ctx, sr, results["decode"].target)
        assert ctx.request.headers[PREFILLER_HEADER]
        assert ctx.request.headers[ENCODER_HEADER]
