"""smg-client — hand-rolled OpenAI-style SDK for the smg gateway
(reference: clients/python/smg_client — SmgClient/AsyncSmgClient, SSE
streaming).

    from smg_client import SmgClient
    client = SmgClient("http://localhost:30000", api_key="...")
    out = client.chat.create(model="m", messages=[...])
    for chunk in client.chat.create(model="m", messages=[...], stream=True): ...
"""
from .client import AsyncSmgClient, SmgClient, SmgError

__all__ = ["AsyncSmgClient", "SmgClient", "SmgError"]
