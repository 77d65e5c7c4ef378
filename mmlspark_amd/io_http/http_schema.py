"""HTTP request/response as structured cells (HTTPSchema.scala parity:
HTTPRequestData:166, HTTPResponseData:90 — full request/response structs that
travel in DataFrame columns)."""
from __future__ import annotations

import json
from dataclasses import dataclass, field
from typing import Dict, Optional


@dataclass
class HTTPRequestData:
    url: str = ""
    method: str = "POST"
    headers: Dict[str, str] = field(default_factory=dict)
    entity: Optional[bytes] = None

    def to_dict(self):
        return {"url": self.url, "method": self.method, "headers": self.headers,
                "entity": self.entity.decode("utf-8", "replace")
                if self.entity else None}


@dataclass
class HTTPResponseData:
    statusCode: int = 200
    reasonPhrase: str = "OK"
    headers: Dict[str, str] = field(default_factory=dict)
    entity: Optional[bytes] = None

    @property
    def text(self) -> str:
        return self.entity.decode("utf-8", "replace") if self.entity else ""

    def json(self):
        return json.loads(self.text) if self.entity else None

    def to_dict(self):
        return {"statusCode": self.statusCode, "reasonPhrase": self.reasonPhrase,
                "headers": self.headers, "entity": self.text}


def string_to_response(s: str, code: int = 200,
                       content_type: str = "text/plain") -> HTTPResponseData:
    return HTTPResponseData(statusCode=code,
                            headers={"Content-Type": content_type},
                            entity=s.encode())


def json_to_response(obj, code: int = 200) -> HTTPResponseData:
    return string_to_response(json.dumps(obj), code, "application/json")
