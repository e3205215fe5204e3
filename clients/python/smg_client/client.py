"""Sync (urllib) and async (aiohttp) SMG clients with SSE streaming."""
from __future__ import annotations

import json
import urllib.request
from typing import Any, Dict, Iterator, List, Optional


class SmgError(RuntimeError):
    def __init__(self, status: int, message: str):
        super().__init__(f"HTTP {status}: {message}")
        self.status = status


def _sse_lines(fp) -> Iterator[dict]:
    for raw in fp:
        line = raw.decode() if isinstance(raw, bytes) else raw
        line = line.strip()
        if line.startswith("data: "):
            data = line[6:]
            if data == "[DONE]":
                return
            try:
                yield json.loads(data)
            except json.JSONDecodeError:
                continue


class _Endpoint:
    def __init__(self, client: "SmgClient", path: str):
        self._client = client
        self._path = path

    def create(self, **kwargs):
        return self._client._post(self._path, kwargs, stream=bool(kwargs.get("stream")))


class SmgClient:
    def __init__(self, base_url: str = "http://127.0.0.1:30000", api_key: Optional[str] = None,
                 timeout: float = 600.0):
        self.base_url = base_url.rstrip("/")
        self.api_key = api_key
        self.timeout = timeout
        self.chat = _Endpoint(self, "/v1/chat/completions")
        self.completions = _Endpoint(self, "/v1/completions")
        self.embeddings = _Endpoint(self, "/v1/embeddings")
        self.responses = _Endpoint(self, "/v1/responses")
        self.messages = _Endpoint(self, "/v1/messages")

    def _headers(self) -> Dict[str, str]:
        h = {"content-type": "application/json"}
        if self.api_key:
            h["authorization"] = f"Bearer {self.api_key}"
        return h

    def _post(self, path: str, body: Dict[str, Any], stream: bool = False):
        req = urllib.request.Request(
            self.base_url + path, data=json.dumps(body).encode(), headers=self._headers(), method="POST"
        )
        try:
            resp = urllib.request.urlopen(req, timeout=self.timeout)
        except urllib.error.HTTPError as e:
            raise SmgError(e.code, e.read().decode()[:500])
        if stream:
            return _sse_lines(resp)
        with resp:
            return json.loads(resp.read())

    def models(self) -> List[dict]:
        with urllib.request.urlopen(self.base_url + "/v1/models", timeout=self.timeout) as resp:
            return json.loads(resp.read())["data"]

    def health(self) -> bool:
        try:
            with urllib.request.urlopen(self.base_url + "/health", timeout=5):
                return True
        except Exception:
            return False


class _AsyncEndpoint:
    def __init__(self, client: "AsyncSmgClient", path: str):
        self._client = client
        self._path = path

    async def create(self, **kwargs):
        if kwargs.get("stream"):
            return self._client._post_stream(self._path, kwargs)
        return await self._client._post(self._path, kwargs)


class AsyncSmgClient:
    def __init__(self, base_url: str = "http://127.0.0.1:30000", api_key: Optional[str] = None,
                 timeout: float = 600.0):
        self.base_url = base_url.rstrip("/")
        self.api_key = api_key
        self.timeout = timeout
        self._session = None
        self.chat = _AsyncEndpoint(self, "/v1/chat/completions")
        self.completions = _AsyncEndpoint(self, "/v1/completions")
        self.embeddings = _AsyncEndpoint(self, "/v1/embeddings")
        self.responses = _AsyncEndpoint(self, "/v1/responses")
        self.messages = _AsyncEndpoint(self, "/v1/messages")

    async def _ensure(self):
        if self._session is None:
            import aiohttp

            headers = {}
            if self.api_key:
                headers["authorization"] = f"Bearer {self.api_key}"
            self._session = aiohttp.ClientSession(
                headers=headers, timeout=aiohttp.ClientTimeout(total=self.timeout)
            )
        return self._session

    async def _post(self, path: str, body: dict):
        session = await self._ensure()
        async with session.post(self.base_url + path, json=body) as resp:
            data = await resp.json()
            if resp.status >= 400:
                raise SmgError(resp.status, json.dumps(data)[:500])
            return data

    async def _post_stream(self, path: str, body: dict):
        session = await self._ensure()
        resp = await session.post(self.base_url + path, json=body)
        if resp.status >= 400:
            raise SmgError(resp.status, (await resp.text())[:500])

        async def gen():
            try:
                async for raw in resp.content:
                    line = raw.decode().strip()
                    if line.startswith("data: "):
                        data = line[6:]
                        if data == "[DONE]":
                            return
                        try:
                            yield json.loads(data)
                        except json.JSONDecodeError:
                            continue
            finally:
                resp.release()

        return gen()

    async def close(self):
        if self._session is not None:
            await self._session.close()
            self._session = None
