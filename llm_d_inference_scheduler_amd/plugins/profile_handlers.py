"""Profile handlers + disaggregation deciders
(parity: pkg/epp/framework/plugins/scheduling/profilehandler/{single,disagg,dataparallel}).

The disagg handler is where P/D/E disaggregation is decided
(disagg_profile_handler.go:246-319 Pick, :323-354 ProcessResults, :360-444
PreRequest): staged picks — decode first, then encode when the encode
decider fires, then prefill when the PD decider fires; decode is primary;
PreRequest publishes `x-prefiller-host-port` / `x-encoder-hosts-ports` so
the node-local execution engine (sidecar equivalent) runs the stages.
"""
from typing import Dict, List, Optional

from ..datalayer.attributes import PREFIX_CACHE_MATCH_INFO
from ..metrics import prom
from ..scheduling.types import ProfileRunResult, SchedulingContext
from .interface import Plugin, ProfileHandler
from .registry import register_plugin

PREFILLER_HEADER = "x-prefiller-host-port"
ENCODER_HEADER = "x-encoder-hosts-ports"
DATA_PARALLEL_HEADER = "x-data-parallel-host-port"


# ---------------- deciders ----------------

class PDDecider(Plugin):
    def should_disaggregate(self, ctx: SchedulingContext,
                            decode_result: ProfileRunResult) -> bool:
        raise NotImplementedError


@register_plugin("prefix-based-pd-decider")
class PrefixBasedPDDecider(PDDecider):
    """Disaggregate when the non-cached suffix exceeds nonCachedTokens
    (disagg/prefix_based_pd_decider.go): consumes PrefixCacheMatchInfo of
    the picked decode endpoint."""

    def __init__(self, name: str = "", **params):
        super().__init__(name, **params)
        self.non_cached_tokens = int(params.get("nonCachedTokens", 512))

    def should_disaggregate(self, ctx, decode_result):
        req = ctx.request
        n_tokens = len(req.prompt_tokens or []) or max(1, req.prompt_len_chars // 4)
        info = ctx.attributes.get(PREFIX_CACHE_MATCH_INFO)
        cached_tokens = 0
        if info is not None and decode_result.target is not None:
            cached_tokens = info.match_blocks.get(
                decode_result.target.name, 0) * info.block_size_tokens
        return (n_tokens - cached_tokens) > self.non_cached_tokens


@register_plugin("always-disagg-pd-decider")
class AlwaysDisaggPDDecider(PDDecider):
    def should_disaggregate(self, ctx, decode_result):
        return True


@register_plugin("always-disagg-multimodal-decider")
class AlwaysDisaggMultimodalDecider(PDDecider):
    """Fires when the request carries multimodal items
    (always_disagg_mm_decider.go: image_url/video_url/input_audio)."""

    def should_disaggregate(self, ctx, decode_result):
        return bool(ctx.request.mm_items)


# ---------------- handlers ----------------

@register_plugin("single-profile-handler")
class SingleProfileHandler(ProfileHandler):
    """Exactly one profile (profilehandler/single)."""

    def __init__(self, name: str = "", **params):
        super().__init__(name, **params)
        self.profile = params.get("profile", "")

    def pick_profiles(self, ctx, profiles, results):
        if results:
            return []
        if self.profile:
            return [self.profile]
        if len(profiles) != 1:
            raise ValueError("single-profile-handler requires exactly one "
                             f"profile, got {sorted(profiles)}")
        return [next(iter(profiles))]

    def process_results(self, ctx, results):
        return next(iter(results)) if results else ""


@register_plugin("disagg-profile-handler", aliases=["pd-profile-handler"])
class DisaggProfileHandler(ProfileHandler):
    """Staged decode -> encode? -> prefill? pick (disagg_profile_handler.go)."""

    def __init__(self, name: str = "", decoders=None, **params):
        super().__init__(name, **params)
        self.decode_profile = params.get("decodeProfile", "decode")
        self.prefill_profile = params.get("prefillProfile", "prefill")
        self.encode_profile = params.get("encodeProfile", "encode")
        self.pd_decider: Optional[PDDecider] = params.get("pdDecider")
        self.encode_decider: Optional[PDDecider] = params.get("encodeDecider")

    def pick_profiles(self, ctx, profiles, results):
        if not results:
            return [self.decode_profile]
        decode_res = results.get(self.decode_profile)
        if decode_res is None or decode_res.target is None:
            return []
        want: List[str] = []
        if (self.encode_profile in profiles
                and self.encode_profile not in results
                and self.encode_decider is not None
                and self.encode_decider.should_disaggregate(ctx, decode_res)):
            want.append(self.encode_profile)
        if (self.prefill_profile in profiles
                and self.prefill_profile not in results
                and self.pd_decider is not None
                and self.pd_decider.should_disaggregate(ctx, decode_res)):
            want.append(self.prefill_profile)
        return want

    def process_results(self, ctx, results):
        decode_res = results.get(self.decode_profile)
        prefill_res = results.get(self.prefill_profile)
        encode_res = results.get(self.encode_profile)
        decision = "decode_only"
        if prefill_res is not None and prefill_res.target is not None:
            decision = "epd" if (encode_res and encode_res.picks) else "pd"
        elif encode_res is not None and encode_res.picks:
            decision = "e_pd"
        prom.disagg_decision_total.labels(decision).inc()
        ctx.state["disagg_decision"] = decision
        return self.decode_profile if decode_res else (
            next(iter(results)) if results else "")

    # PreRequest hook (disagg_profile_handler.go:360-444): publish stage
    # targets as headers on the outgoing request.
    def pre_request(self, ctx: SchedulingContext, result, target) -> None:
        prefill_res = result.profile_results.get(self.prefill_profile)
        if prefill_res is not None and prefill_res.target is not None:
            ctx.request.headers[PREFILLER_HEADER] = \
                prefill_res.target.metadata.address
        encode_res = result.profile_results.get(self.encode_profile)
        if encode_res is not None and encode_res.picks:
            ctx.request.headers[ENCODER_HEADER] = ",".join(
                ep.metadata.address for ep in encode_res.picks)


@register_plugin("data-parallel-profile-handler")
class DataParallelProfileHandler(SingleProfileHandler):
    """Deprecated DP handler (dataparallel/dp_profile_handler.go:19-26):
    single profile + `x-data-parallel-host-port` header rewrite."""

    def __init__(self, name: str = "", **params):
        super().__init__(name, **params)
        self.primary_port = params.get("primaryPort", "")

    def pre_request(self, ctx, result, target) -> None:
        if target is None:
            return
        ctx.request.headers[DATA_PARALLEL_HEADER] = target.metadata.address
