"""OpenAI Files API backing store (reference services/files_service parity:
local-disk storage at <root>/<user>/<file_id>)."""

from __future__ import annotations

import json
import os
import time
import uuid
from dataclasses import asdict, dataclass
from typing import Dict, List, Optional


@dataclass
class OpenAIFile:
    id: str
    bytes: int
    created_at: int
    filename: str
    object: str = "file"
    purpose: str = "batch"

    def metadata(self) -> Dict:
        return asdict(self)


class FileStorage:
    def __init__(self, root: str = "/tmp/vllm_files") -> None:
        self.root = root
        os.makedirs(root, exist_ok=True)

    def _user_dir(self, user: str) -> str:
        d = os.path.join(self.root, user or "anonymous")
        os.makedirs(d, exist_ok=True)
        return d

    def save_file(
        self,
        content: bytes,
        filename: str,
        purpose: str = "batch",
        user: str = "anonymous",
    ) -> OpenAIFile:
        file_id = f"file-{uuid.uuid4().hex[:24]}"
        d = self._user_dir(user)
        with open(os.path.join(d, file_id), "wb") as f:
            f.write(content)
        meta = OpenAIFile(
            id=file_id,
            bytes=len(content),
            created_at=int(time.time()),
            filename=filename,
            purpose=purpose,
        )
        with open(os.path.join(d, file_id + ".json"), "w") as f:
            json.dump(meta.metadata(), f)
        return meta

    def get_file_metadata(
        self, file_id: str, user: str = "anonymous"
    ) -> Optional[OpenAIFile]:
        path = os.path.join(self._user_dir(user), file_id + ".json")
        if not os.path.exists(path):
            return None
        with open(path) as f:
            return OpenAIFile(**json.load(f))

    def get_file_content(
        self, file_id: str, user: str = "anonymous"
    ) -> Optional[bytes]:
        path = os.path.join(self._user_dir(user), file_id)
        if not os.path.exists(path):
            return None
        with open(path, "rb") as f:
            return f.read()

    def list_files(self, user: str = "anonymous") -> List[OpenAIFile]:
        d = self._user_dir(user)
        out = []
        for name in os.listdir(d):
            if name.endswith(".json"):
                with open(os.path.join(d, name)) as f:
                    out.append(OpenAIFile(**json.load(f)))
        return out

    def delete_file(self, file_id: str, user: str = "anonymous") -> bool:
        d = self._user_dir(user)
        found = False
        for suffix in ("", ".json"):
            p = os.path.join(d, file_id + suffix)
            if os.path.exists(p):
                os.remove(p)
                found = True
        return found
