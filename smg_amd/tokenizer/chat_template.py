"""Chat-template rendering (reference: crates/tokenizer/src/chat_template.rs —
minijinja; here jinja2 with the same HF template semantics)."""
from __future__ import annotations

from typing import Any, Dict, List, Optional

DEFAULT_TEMPLATE = (
    "{% for message in messages %}"
    "<|{{ message.role }}|>\n{{ message.content }}\n"
    "{% endfor %}"
    "{% if add_generation_prompt %}<|assistant|>\n{% endif %}"
)


class ChatTemplate:
    def __init__(self, template: Optional[str] = None):
        import jinja2

        self.source = template or DEFAULT_TEMPLATE
        env = jinja2.Environment(autoescape=False, trim_blocks=True, lstrip_blocks=True)
        env.globals["raise_exception"] = self._raise
        self._tpl = env.from_string(self.source)

    @staticmethod
    def _raise(msg):
        raise ValueError(msg)

    @staticmethod
    def _normalize(messages: List[Dict[str, Any]]) -> List[Dict[str, Any]]:
        out = []
        for m in messages:
            content = m.get("content")
            if isinstance(content, list):  # multimodal parts -> text parts joined
                content = "".join(
                    p.get("text", "") for p in content if isinstance(p, dict) and p.get("type") == "text"
                )
            out.append({**m, "content": content or ""})
        return out

    def render(
        self,
        messages: List[Dict[str, Any]],
        add_generation_prompt: bool = True,
        tools: Optional[List[dict]] = None,
        **extra,
    ) -> str:
        return self._tpl.render(
            messages=self._normalize(messages),
            add_generation_prompt=add_generation_prompt,
            tools=tools,
            bos_token="",
            eos_token="",
            **extra,
        )


def load_chat_template(path_or_inline: Optional[str]) -> ChatTemplate:
    if not path_or_inline:
        return ChatTemplate()
    import os

    if os.path.exists(path_or_inline):
        with open(path_or_inline) as f:
            return ChatTemplate(f.read())
    return ChatTemplate(path_or_inline)
