"""LocalEngineClient: the OpenAI-client-shaped handle over the local engine.

The reference constructs ``openai.OpenAI`` and calls
``client.chat.completions.create`` / ``client.beta.chat.completions.parse`` /
``client.embeddings.create`` (k_llms/client.py:34, completions.py:73,134).
This class exposes exactly those surfaces, backed by LLMEngine: one call =
one batched local generation (shared prefill, n fanned decode streams).
"""

from __future__ import annotations

import threading
import time
import uuid
from typing import Any, Dict, List, Optional

import asyncio

from pydantic import BaseModel

from ..types.openai_compat import (
    ChatCompletion,
    ChatCompletionMessage,
    ChatCompletionTokenLogprob,
    TopLogprob,
    Choice,
    ChoiceLogprobs,
    CompletionUsage,
    CreateEmbeddingResponse,
    Embedding,
    EmbeddingUsage,
    ParsedChatCompletion,
    ParsedChatCompletionMessage,
    ParsedChoice,
)
from .config import EngineConfig
from .sampling import SamplingParams


def _parse_logit_bias(lb):
    """OpenAI logit_bias {token_id: bias}; malformed keys/values raise a
    clean ValueError (a 400), not a TypeError."""
    if not lb:
        return None
    try:
        return {int(k): float(v) for k, v in lb.items()}
    except (TypeError, ValueError, AttributeError):
        raise ValueError("logit_bias must map token ids to numeric biases")


# OpenAI responses carry a backend build fingerprint; ours names the local
# engine so response provenance is visible in logs/dumps
from .. import __version__ as _pkg_version

_FINGERPRINT = f"kllms_amd-{_pkg_version}-gfx950"


class ContextLengthExceededError(ValueError):
    """Prompt does not fit in the model's context window (OpenAI-compatible
    behavior: the API raises context_length_exceeded rather than silently
    truncating the prompt, which would cut the chat template's assistant
    header and make the model continue the user text)."""


class LocalEngineClient:
    """Engine handle with .chat/.beta/.embeddings namespaces."""

    def __init__(self, **config_kwargs: Any):
        engine = config_kwargs.pop("llm_engine", None)
        self.config = engine.config if engine is not None else EngineConfig(**config_kwargs)
        self._engine = engine
        self._engine_lock = threading.Lock()
        self.chat = _ChatNS(self)
        self.beta = _BetaNS(self)
        self.embeddings = _EmbeddingsNS(self)
        # model used by the llm string-consensus mode: the local
        # serving model itself.
        self.consensus_model = self.config.model

    # --- engine lifecycle -----------------------------------------------------
    @property
    def scheduler(self):
        """Continuous-batching scheduler (lazy): concurrent submissions merge
        into one running decode batch. Shares the engine lock with the direct
        path so both can be used."""
        if getattr(self, "_scheduler", None) is None:
            from .scheduler import BatchScheduler

            self._scheduler = BatchScheduler(self.engine, engine_lock=self._engine_lock)
        return self._scheduler

    @property
    def engine(self):
        if self._engine is None:
            with self._engine_lock:
                if self._engine is None:
                    from .engine import LLMEngine

                    self._engine = LLMEngine(self.config)
        return self._engine

    @property
    def tokenizer(self):
        return self.engine.tokenizer

    def crop_to_tokens(self, text: str, max_tokens: int) -> str:
        return self.tokenizer.crop_to_tokens(text, max_tokens)

    # --- core generation ------------------------------------------------------
    @staticmethod
    def _validate_call_params(call_params: Dict[str, Any]) -> None:
        """OpenAI-style 400s for out-of-range sampling params (the reference
        relies on the remote API to raise these; locally we mirror the
        documented ranges so a switched-over client sees the same errors)."""
        def num(key):
            v = call_params.get(key)
            if v is None:
                return None
            try:
                return float(v) if not isinstance(v, bool) else 1.0 * v
            except (TypeError, ValueError):
                raise ValueError(f"{key} must be a number, got {type(v).__name__}")

        t = num("temperature")
        if t is not None and not 0.0 <= t <= 2.0:
            raise ValueError(f"temperature must be between 0 and 2, got {t}")
        p = num("top_p")
        if p is not None and not 0.0 <= p <= 1.0:
            raise ValueError(f"top_p must be between 0 and 1, got {p}")
        n = num("n")
        if n is not None and not 1 <= n <= 128:
            raise ValueError(f"n must be between 1 and 128, got {n}")
        mt = num("max_tokens")
        if mt is not None and mt < 1:
            raise ValueError(f"max_tokens must be at least 1, got {mt}")
        for key in ("frequency_penalty", "presence_penalty"):
            v = num(key)
            if v is not None and not -2.0 <= v <= 2.0:
                raise ValueError(f"{key} must be between -2 and 2, got {v}")
        for key in ("seed", "top_logprobs"):
            v = call_params.get(key)
            if v is not None and not isinstance(v, int):
                raise ValueError(f"{key} must be an integer, got {type(v).__name__}")
        lb = call_params.get("logit_bias")
        if lb is not None and not isinstance(lb, dict):
            raise ValueError(f"logit_bias must be an object, got {type(lb).__name__}")
        msgs = call_params.get("messages")
        if msgs is not None:
            for m in msgs:
                if not isinstance(m, dict) or not isinstance(m.get("role", ""), str):
                    raise ValueError("each message must be an object with a string 'role'")
                c = m.get("content")
                if c is None or isinstance(c, str):
                    continue
                if isinstance(c, list):
                    # OpenAI content-part arrays: text parts are supported
                    # (flattened before templating); image/audio parts are not
                    for part in c:
                        if not isinstance(part, dict) or part.get("type") != "text" \
                                or not isinstance(part.get("text"), str):
                            raise ValueError(
                                "only text content parts are supported by the local engine")
                    continue
                raise ValueError("message 'content' must be a string, part list or null")

    @staticmethod
    def _flatten_messages(messages: List[Dict[str, Any]]) -> List[Dict[str, Any]]:
        """Flatten OpenAI content-part arrays to plain strings for the chat
        template (validation guarantees text-only parts)."""
        out = []
        for m in messages:
            c = m.get("content")
            if isinstance(c, list):
                m = {**m, "content": "".join(p["text"] for p in c)}
            out.append(m)
        return out

    def _generate(self, call_params: Dict[str, Any], constrained: bool, scheduled: bool = False) -> tuple:
        from .engine import GenRequest

        messages: List[Dict[str, Any]] = call_params["messages"]
        if not messages:
            raise ValueError("messages must be a non-empty list")
        self._validate_call_params(call_params)
        model: str = call_params.get("model", self.config.model)
        n: int = int(call_params.get("n") or 1)

        sampling = SamplingParams(
            temperature=call_params.get("temperature", 1.0),
            top_p=call_params.get("top_p", 1.0),
            top_k=call_params.get("top_k", 0),
            max_tokens=call_params.get("max_tokens"),
            stop=call_params.get("stop"),
            seed=call_params.get("seed"),
            frequency_penalty=call_params.get("frequency_penalty", 0.0),
            presence_penalty=call_params.get("presence_penalty", 0.0),
            logprobs=bool(call_params.get("logprobs", False)),
            top_logprobs=int(call_params.get("top_logprobs") or 0),
            logit_bias=_parse_logit_bias(call_params.get("logit_bias")),
        )
        if sampling.top_logprobs and not sampling.logprobs:
            raise ValueError("top_logprobs requires logprobs=True")
        if not 0 <= sampling.top_logprobs <= 20:
            raise ValueError("top_logprobs must be between 0 and 20")

        constraint = None
        tool_name = None
        response_format = call_params.get("response_format")
        if response_format is not None:
            constraint = self._build_constraint(response_format, constrained)
        elif call_params.get("tools"):
            constraint, tool_name = self._build_tool_constraint(call_params)

        eng = self.engine
        prompt = eng.tokenizer.apply_chat_template(self._flatten_messages(messages))
        prompt_ids = eng.tokenizer.encode(prompt)
        self._fit_context(prompt_ids, sampling)

        tmo = call_params.get("timeout")
        deadline = (time.monotonic() + float(tmo)) if tmo else None
        req = GenRequest(prompt_ids=prompt_ids, n=n, sampling=sampling, constraint=constraint,
                         deadline=deadline)
        if scheduled:
            out = self.scheduler.submit(req).result()
        else:
            with self._engine_lock:
                out = eng.generate([req])[0]
        return out, model, sampling, tool_name

    def _fit_context(self, prompt_ids: List[int], sampling: SamplingParams) -> None:
        """Error on over-long prompts (never truncate — that would cut the
        chat template's assistant header); clamp max_tokens to the remaining
        window when the prompt itself fits."""
        max_seq = self.config.max_seq_len
        if len(prompt_ids) >= max_seq:
            raise ContextLengthExceededError(
                f"prompt is {len(prompt_ids)} tokens but the context window is "
                f"{max_seq} tokens (context_length_exceeded)"
            )
        room = max_seq - len(prompt_ids)
        want = sampling.max_tokens or self.config.default_max_new_tokens
        sampling.max_tokens = min(want, room)

    def _build_tool_constraint(self, call_params: Dict[str, Any]):
        """FORCED tool calls served natively: with a specific function forced
        via tool_choice (or tool_choice="required" and exactly one tool), the
        function's JSON-schema parameters compile into the decoding DFA and
        the output is returned as an OpenAI tool_calls message. tool_choice
        "auto" is NOT model-decided locally — it generates plain content
        (documented limitation; the reference delegates the decision to the
        remote model)."""
        from .constrained import JsonSchemaConstraint

        tools = call_params.get("tools") or []
        choice = call_params.get("tool_choice")
        fn = None
        if isinstance(choice, dict) and choice.get("type") == "function":
            name = (choice.get("function") or {}).get("name")
            for t in tools:
                f = t.get("function", {})
                if f.get("name") == name:
                    fn = f
                    break
            if fn is None:
                raise ValueError(f"tool_choice names unknown function {name!r}")
        elif choice == "required" and len(tools) == 1:
            fn = tools[0].get("function", {})
        if fn is None:
            return None, None
        schema = fn.get("parameters") or {"type": "object"}
        return (
            JsonSchemaConstraint(schema, self.engine.tokenizer,
                                 whitespace=getattr(self.config, "constrained_whitespace", False)),
            fn.get("name"),
        )

    def _build_constraint(self, response_format: Any, constrained: bool):
        from .constrained import JsonSchemaConstraint

        schema = None
        if isinstance(response_format, type) and issubclass(response_format, BaseModel):
            schema = response_format.model_json_schema()
        elif isinstance(response_format, dict):
            t = response_format.get("type")
            if t == "json_schema":
                schema = response_format.get("json_schema", {}).get("schema")
            elif t == "json_object":
                schema = {"type": "object"}
        if schema is None:
            return None
        try:
            return JsonSchemaConstraint(
                schema, self.engine.tokenizer,
                whitespace=getattr(self.config, "constrained_whitespace", False),
            )
        except Exception:
            if constrained:
                raise
            return None

    def _mk_logprobs(self, stream) -> Optional[ChoiceLogprobs]:
        toks = []
        alt_lists = getattr(stream, "top_logprobs", []) or []
        for pos, (tid, lp) in enumerate(zip(stream.token_ids, stream.logprobs)):
            s = self.engine.tokenizer.decode([tid])
            tops = []
            if pos < len(alt_lists):
                for a_tid, a_lp in alt_lists[pos]:
                    a_s = self.engine.tokenizer.decode([a_tid])
                    tops.append(TopLogprob(token=a_s, bytes=list(a_s.encode()), logprob=a_lp))
            toks.append(
                ChatCompletionTokenLogprob(token=s, bytes=list(s.encode()), logprob=lp, top_logprobs=tops)
            )
        return ChoiceLogprobs(content=toks)

    def _mk_timings(self, out) -> Dict[str, Any]:
        """Per-request serving metrics (tokens/s, per-phase wall
        times) attached as an extra `timings` field on the response."""
        tm = dict(getattr(self.engine, "last_timings", {}))
        completion_tokens = sum(len(s.token_ids) for s in out.streams)
        decode_s = tm.get("decode_ms", 0.0) / 1000.0
        tm["prefill_ms_request"] = out.prefill_ms
        tm["decode_ms_request"] = out.decode_ms
        if decode_s > 0:
            tm["decode_tokens_per_s_batch"] = round(
                tm.get("decode_steps", 0) * tm.get("n_streams", 0) / decode_s, 2
            )
        tm["completion_tokens"] = completion_tokens
        return tm

    def _mk_usage(self, out) -> CompletionUsage:
        completion_tokens = sum(len(s.token_ids) for s in out.streams)
        # prompt charged once for n>1 — shared prefill (OpenAI-matching
        # semantics)
        return CompletionUsage(
            prompt_tokens=out.prompt_tokens,
            completion_tokens=completion_tokens,
            total_tokens=out.prompt_tokens + completion_tokens,
        )

    # --- public: plain completions ---------------------------------------------
    def chat_completions_create_many(self, call_params_list: List[Dict[str, Any]]) -> List[ChatCompletion]:
        """Batched serving path: ALL requests go through ONE engine.generate()
        call — every prompt prefilled in one packed varlen batch, every decode
        stream stepped together. Used by the bench/serving layer."""
        from .engine import GenRequest

        eng = self.engine
        reqs = []
        samplings = []
        for call_params in call_params_list:
            messages = call_params["messages"]
            self._validate_call_params(call_params)
            n = int(call_params.get("n") or 1)
            sampling = SamplingParams(
                temperature=call_params.get("temperature", 1.0),
                top_p=call_params.get("top_p", 1.0),
                top_k=call_params.get("top_k", 0),
                max_tokens=call_params.get("max_tokens"),
                stop=call_params.get("stop"),
                seed=call_params.get("seed"),
                frequency_penalty=call_params.get("frequency_penalty", 0.0),
                presence_penalty=call_params.get("presence_penalty", 0.0),
                logprobs=bool(call_params.get("logprobs", False)),
                top_logprobs=int(call_params.get("top_logprobs") or 0),
                logit_bias=_parse_logit_bias(call_params.get("logit_bias")),
            )
            constraint = None
            rf = call_params.get("response_format")
            if rf is not None:
                constraint = self._build_constraint(rf, constrained=False)
            prompt = eng.tokenizer.apply_chat_template(self._flatten_messages(messages))
            prompt_ids = eng.tokenizer.encode(prompt)
            self._fit_context(prompt_ids, sampling)
            reqs.append(GenRequest(prompt_ids=prompt_ids, n=n, sampling=sampling, constraint=constraint))
            samplings.append(sampling)
        with self._engine_lock:
            outs = eng.generate(reqs)
        results = []
        for call_params, out, sampling in zip(call_params_list, outs, samplings):
            choices = [
                Choice(
                    finish_reason=s.finish_reason,
                    index=i,
                    message=ChatCompletionMessage(role="assistant", content=s.text),
                    logprobs=self._mk_logprobs(s) if sampling.logprobs else None,
                )
                for i, s in enumerate(out.streams)
            ]
            results.append(
                ChatCompletion(
                    id=f"chatcmpl-{uuid.uuid4().hex[:24]}",
                    choices=choices,
                    created=int(time.time()),
            system_fingerprint=_FINGERPRINT,
                    model=call_params.get("model", self.config.model),
                    usage=self._mk_usage(out),
                )
            )
        return results

    def chat_completions_create(self, _scheduled: bool = False, **call_params: Any) -> ChatCompletion:
        out, model, sampling, tool_name = self._generate(call_params, constrained=False, scheduled=_scheduled)
        choices = []
        for i, s in enumerate(out.streams):
            if tool_name is not None:
                from ..types.openai_compat import ChatCompletionMessageToolCall, Function

                msg = ChatCompletionMessage(
                    role="assistant", content=None,
                    tool_calls=[ChatCompletionMessageToolCall(
                        id=f"call_{uuid.uuid4().hex[:24]}",
                        function=Function(name=tool_name, arguments=s.text))],
                )
                fr = "tool_calls" if s.finish_reason == "stop" else s.finish_reason
            else:
                msg = ChatCompletionMessage(role="assistant", content=s.text)
                fr = s.finish_reason
            choices.append(
                Choice(
                    finish_reason=fr,
                    index=i,
                    message=msg,
                    logprobs=self._mk_logprobs(s) if sampling.logprobs else None,
                )
            )
        return ChatCompletion(
            id=f"chatcmpl-{uuid.uuid4().hex[:24]}",
            choices=choices,
            created=int(time.time()),
            system_fingerprint=_FINGERPRINT,
            model=model,
            usage=self._mk_usage(out),
            timings=self._mk_timings(out),
        )

    def chat_completions_parse(self, _scheduled: bool = False, **call_params: Any) -> ParsedChatCompletion:
        import json

        response_format = call_params.get("response_format")
        out, model, sampling, _ = self._generate(call_params, constrained=True, scheduled=_scheduled)
        choices = []
        for i, s in enumerate(out.streams):
            parsed = None
            if isinstance(response_format, type) and issubclass(response_format, BaseModel):
                try:
                    parsed = response_format.model_validate(json.loads(s.text))
                except Exception:
                    parsed = None
            choices.append(
                ParsedChoice(
                    finish_reason=s.finish_reason,
                    index=i,
                    message=ParsedChatCompletionMessage(role="assistant", content=s.text, parsed=parsed),
                    logprobs=self._mk_logprobs(s) if sampling.logprobs else None,
                )
            )
        return ParsedChatCompletion(
            id=f"chatcmpl-{uuid.uuid4().hex[:24]}",
            choices=choices,
            created=int(time.time()),
            system_fingerprint=_FINGERPRINT,
            model=model,
            usage=self._mk_usage(out),
            timings=self._mk_timings(out),
        )

    # --- embeddings -------------------------------------------------------------
    def embeddings_create(self, input: List[str], model: str = "text-embedding-3-small") -> CreateEmbeddingResponse:
        eng = self.engine  # materialize outside the lock (lock is not reentrant)
        with self._engine_lock:
            vecs, total_tokens = eng.embed(list(input))
        data = [Embedding(embedding=v, index=i) for i, v in enumerate(vecs)]
        return CreateEmbeddingResponse(
            data=data, model=model, usage=EmbeddingUsage(prompt_tokens=total_tokens, total_tokens=total_tokens)
        )


class _CompletionsNS:
    def __init__(self, client: LocalEngineClient):
        self._client = client

    def create(self, **kw) -> ChatCompletion:
        return self._client.chat_completions_create(**kw)

    async def acreate(self, **kw) -> ChatCompletion:
        # async requests go through the continuous-batching scheduler so
        # concurrent awaits merge into one decode batch
        return await asyncio.to_thread(self._client.chat_completions_create, True, **kw)

    def parse(self, **kw) -> ParsedChatCompletion:
        return self._client.chat_completions_parse(**kw)

    async def aparse(self, **kw) -> ParsedChatCompletion:
        return await asyncio.to_thread(self._client.chat_completions_parse, True, **kw)


class _ChatNS:
    def __init__(self, client: LocalEngineClient):
        self.completions = _CompletionsNS(client)


class _BetaNS:
    def __init__(self, client: LocalEngineClient):
        self.chat = _ChatNS(client)


class _EmbeddingsNS:
    def __init__(self, client: LocalEngineClient):
        self._client = client

    def create(self, input: List[str], model: str = "text-embedding-3-small") -> CreateEmbeddingResponse:
        return self._client.embeddings_create(input=input, model=model)
