"""Jinja chat templating (reference: chat_template/ over vendored minja —
SURVEY.md 2.10; here on real jinja2, which is a superset of the minja
subset).

Template resolution order: explicit template string > chat_template.jinja /
tokenizer_config.json in the model dir > the built-in ChatML default.
Supports tools, chat_template_kwargs and tool_choice=="none" suppression.
"""
from __future__ import annotations

import json
import os
from typing import Any, Dict, List, Optional

import jinja2

DEFAULT_CHATML = (
    "{% for message in messages %}"
    "{{ '<|im_start|>' + message['role'] + '\n' }}"
    "{% if message['content'] is string %}{{ message['content'] }}"
    "{% else %}"
    "{% for part in message['content'] %}"
    "{% if part['type'] == 'text' %}{{ part['text'] }}{% endif %}"
    "{% endfor %}"
    "{% endif %}{{ '<|im_end|>\n' }}"
    "{% endfor %}"
    "{% if add_generation_prompt %}{{ '<|im_start|>assistant\n' }}{% endif %}"
)


class JinjaChatTemplate:
    def __init__(self, template: Optional[str] = None,
                 model_dir: Optional[str] = None):
        if template is None and model_dir:
            template = self._load_from_dir(model_dir)
        self.source = template or DEFAULT_CHATML
        env = jinja2.Environment(  # noqa: S701 — text templating, not HTML
            loader=jinja2.BaseLoader(), trim_blocks=True, lstrip_blocks=True)
        env.globals["raise_exception"] = self._raise
        env.filters["tojson"] = lambda x, **kw: json.dumps(x, **kw)
        self.template = env.from_string(self.source)

    @staticmethod
    def _raise(msg):
        raise jinja2.TemplateError(msg)

    @staticmethod
    def _load_from_dir(model_dir: str) -> Optional[str]:
        p = os.path.join(model_dir, "chat_template.jinja")
        if os.path.exists(p):
            with open(p) as f:
                return f.read()
        p = os.path.join(model_dir, "tokenizer_config.json")
        if os.path.exists(p):
            with open(p) as f:
                cfg = json.load(f)
            ct = cfg.get("chat_template")
            if isinstance(ct, list):  # multi-template form
                for entry in ct:
                    if entry.get("name") == "default":
                        return entry.get("template")
                return ct[0].get("template") if ct else None
            return ct
        return None

    def apply(self, messages: List[Dict[str, Any]],
              tools: Optional[List[Dict[str, Any]]] = None,
              tool_choice: Optional[Any] = None,
              add_generation_prompt: bool = True,
              chat_template_kwargs: Optional[Dict[str, Any]] = None) -> str:
        # tool_choice == "none": render without tool definitions
        # (reference: Scheduler::schedule honouring tool_choice)
        if tool_choice == "none":
            tools = None
        ctx = dict(messages=messages, tools=tools,
                   add_generation_prompt=add_generation_prompt)
        ctx.update(chat_template_kwargs or {})
        return self.template.render(**ctx)
