"""Chat-template registry (re-implementation of the reference's
cmd/tuning/template.py registry: 17+ named templates, multiturn encode
with bos/eos handling, source/target pair output for -100 masking).

A template turns (query, response, history, system) into a list of
(source_ids, target_ids) pairs; the dataset layer masks source_ids with
IGNORE_INDEX (template.py:92-119 behavior).
"""

from __future__ import annotations

from dataclasses import dataclass, field
from typing import List, Optional, Tuple

TEMPLATES = {}


def _tok_encode(tokenizer, text: str) -> List[int]:
    """Encode without special tokens regardless of tokenizer flavor."""
    if hasattr(tokenizer, "encode"):
        try:
            return tokenizer.encode(text, add_special_tokens=False)
        except TypeError:
            return tokenizer.encode(text)
    raise TypeError("tokenizer must expose .encode")


@dataclass
class Template:
    prefix: List[str] = field(default_factory=list)   # system preamble
    prompt: List[str] = field(default_factory=lambda: ["{{query}}"])
    sep: List[str] = field(default_factory=list)
    stop_words: List[str] = field(default_factory=list)
    use_history: bool = True
    efficient_eos: bool = False

    def _render(self, parts: List[str], query: str = "", idx: str = "",
                system: str = "") -> str:
        out = []
        for p in parts:
            out.append(p.replace("{{system}}", system)
                        .replace("{{query}}", query)
                        .replace("{{idx}}", idx))
        return "".join(out)

    def encode_oneturn(self, tokenizer, query: str, resp: str,
                       history=None, system: str = ""):
        pairs = self.encode_multiturn(tokenizer, query, resp, history, system)
        src: List[int] = []
        for s, t in pairs[:-1]:
            src.extend(s)
            src.extend(t)
        src.extend(pairs[-1][0])
        return src, pairs[-1][1]

    def encode_multiturn(self, tokenizer, query: str, resp: str,
                         history: Optional[List[Tuple[str, str]]] = None,
                         system: str = "") -> List[Tuple[List[int], List[int]]]:
        history = list(history or []) if self.use_history else []
        history = history + [(query, resp)]
        bos = ([tokenizer.bos_token_id]
               if getattr(tokenizer, "bos_token_id", None) is not None else [])
        eos = ([tokenizer.eos_token_id]
               if getattr(tokenizer, "eos_token_id", None) is not None else [])
        sep_text = self._render(self.sep)
        pairs = []
        for i, (q, r) in enumerate(history):
            if i == 0:
                text = self._render(self.prefix, system=system) + \
                    self._render(self.prompt, query=q, idx=str(i + 1),
                                 system=system)
                src = bos + _tok_encode(tokenizer, text)
            else:
                text = sep_text + self._render(self.prompt, query=q,
                                               idx=str(i + 1), system=system)
                src = _tok_encode(tokenizer, text)
            tgt = _tok_encode(tokenizer, r) + eos
            pairs.append((src, tgt))
        return pairs


@dataclass
class Llama2Template(Template):
    """Llama-2 [INST] convention: system folded into the first turn,
    bos per turn (template.py:154-173 behavior)."""

    def encode_multiturn(self, tokenizer, query, resp, history=None,
                         system: str = ""):
        history = list(history or []) if self.use_history else []
        history = history + [(query, resp)]
        bos = ([tokenizer.bos_token_id]
               if getattr(tokenizer, "bos_token_id", None) is not None else [])
        eos = ([tokenizer.eos_token_id]
               if getattr(tokenizer, "eos_token_id", None) is not None else [])
        sys_text = self._render(self.prefix, system=system) if system else ""
        pairs = []
        for i, (q, r) in enumerate(history):
            q_text = sys_text + q if i == 0 else q
            text = self._render(self.prompt, query=q_text)
            pairs.append((bos + _tok_encode(tokenizer, text),
                          _tok_encode(tokenizer, r) + eos))
        return pairs


def register_template(name: str, template: Template):
    TEMPLATES[name] = template
    return template


def get_template(name: str) -> Template:
    if name not in TEMPLATES:
        raise KeyError(f"unknown template {name!r}; have {sorted(TEMPLATES)}")
    return TEMPLATES[name]


def get_template_and_fix_tokenizer(name: str, tokenizer):
    """Parity with template.py:201-222: ensure eos/pad exist."""
    t = get_template(name)
    if getattr(tokenizer, "eos_token_id", None) is None and \
            hasattr(tokenizer, "add_special_tokens"):
        tokenizer.add_special_tokens({"eos_token": "</s>"})
    if getattr(tokenizer, "pad_token_id", None) is None:
        if hasattr(tokenizer, "pad_token"):
            tokenizer.pad_token = getattr(tokenizer, "eos_token", "</s>")
    return t


_LLAMA2_SYS = "<<SYS>>\n{{system}}\n<</SYS>>\n\n"

register_template("vanilla", Template(prompt=["{{query}}"], use_history=False))
register_template("default", Template(
    prefix=["{{system}}"],
    prompt=["Human: {{query}}\nAssistant: "], sep=["\n"]))
register_template("llama2", Llama2Template(
    prefix=[_LLAMA2_SYS], prompt=["[INST] {{query}} [/INST] "]))
register_template("llama2_zh", Llama2Template(
    prefix=[_LLAMA2_SYS], prompt=["[INST] {{query}} [/INST] "]))
register_template("alpaca", Template(
    prefix=["{{system}}\n\n"],
    prompt=["### Instruction:\n{{query}}\n\n### Response:\n"], sep=["\n\n"]))
register_template("vicuna", Template(
    prefix=["{{system}} "],
    prompt=["USER: {{query}} ASSISTANT: "]))
register_template("belle", Template(
    prompt=["Human: {{query}}\n\nBelle: "], sep=["\n\n"]))
register_template("ziya", Template(
    prompt=["<human>:{{query}}\n<bot>:"], sep=["\n"]))
register_template("aquila", Template(
    prefix=["{{system}}"],
    prompt=["Human: {{query}}###Assistant: "], sep=["###"]))
register_template("intern", Template(
    prompt=["<|User|>:{{query}}<eoh>\n<|Bot|>:"], sep=["<eoa>\n"],
    stop_words=["<eoa>"]))
register_template("baichuan", Template(
    prompt=["<reserved_102>{{query}}<reserved_103>"]))
register_template("baichuan2", Template(
    prompt=["<reserved_106>{{query}}<reserved_107>"]))
register_template("starchat", Template(
    prefix=["<|system|>\n{{system}}<|end|>\n"],
    prompt=["<|user|>\n{{query}}<|end|>\n<|assistant|>\n"], sep=["<|end|>\n"],
    stop_words=["<|end|>"]))
register_template("chatml", Template(
    prefix=["<|im_start|>system\n{{system}}<|im_end|>\n"],
    prompt=["<|im_start|>user\n{{query}}<|im_end|>\n<|im_start|>assistant\n"],
    sep=["<|im_end|>\n"], stop_words=["<|im_end|>"]))
register_template("chatglm2", Template(
    prompt=["[Round {{idx}}]\n\n问：{{query}}\n\n答："], sep=["\n\n"]))
register_template("chatglm3", Template(
    prefix=["<|system|>\n{{system}}"],
    prompt=["<|user|>\n{{query}}<|assistant|>\n"]))
@dataclass
class Llama3Template(Template):
    """Llama-3 header convention: the system block is omitted entirely
    when no system prompt is given, and assistant turns terminate with
    <|eot_id|> (the turn terminator the engine's stop set knows) rather
    than the tokenizer's plain eos."""

    def encode_multiturn(self, tokenizer, query, resp, history=None,
                         system: str = ""):
        if not system:
            # non-mutating copy: templates are shared module singletons
            # and the serving front encodes from multiple threads
            import dataclasses
            pairs = Template.encode_multiturn(
                dataclasses.replace(self, prefix=[]), tokenizer, query,
                resp, history, system)
        else:
            pairs = super().encode_multiturn(tokenizer, query, resp,
                                             history, system)
        eos = getattr(tokenizer, "eos_token_id", None)
        eot = _tok_encode(tokenizer, "<|eot_id|>")
        out = []
        for src, tgt in pairs:
            if eos is not None and tgt and tgt[-1] == eos:
                tgt = tgt[:-1] + eot
            out.append((src, tgt))
        return out


register_template("llama3", Llama3Template(
    prefix=["<|start_header_id|>system<|end_header_id|>\n\n"
            "{{system}}<|eot_id|>"],
    prompt=["<|start_header_id|>user<|end_header_id|>\n\n{{query}}<|eot_id|>"
            "<|start_header_id|>assistant<|end_header_id|>\n\n"],
    stop_words=["<|eot_id|>"]))
register_template("openchat", Template(
    prompt=["GPT4 User: {{query}}<|end_of_turn|>GPT4 Assistant: "],
    sep=["<|end_of_turn|>"], stop_words=["<|end_of_turn|>"]))
register_template("xverse", Template(
    prefix=["{{system}}"],
    prompt=["Human: {{query}}\n\nAssistant: "]))
