"""GitHub App authentication (reference: py/code_intelligence/github_app.py).

GitHubApp: JWT (RS256, 60 s) -> installation id (cached) -> installation
access token; FixedAccessTokenGenerator resolves a PAT from the same env
var chain as the reference (github_app.py:265-303); GitHubAppTokenGenerator
auto-refreshes expired installation tokens (github_app.py:305-364).
HTTP session injectable for offline tests."""
from __future__ import annotations

import datetime
import logging
import os
from typing import Dict, Optional

from .rs256 import app_jwt

log = logging.getLogger(__name__)

GITHUB_API = "https://api.github.com"


class GitHubApp:
    def __init__(self, pem_path: Optional[str] = None, app_id: Optional[str] = None,
                 pem_contents: Optional[str] = None, session=None,
                 api_url: str = GITHUB_API):
        if pem_contents is None:
            if pem_path is None:
                raise ValueError("need pem_path or pem_contents")
            with open(pem_path) as f:
                pem_contents = f.read()
        self.pem = pem_contents
        self.app_id = str(app_id)
        self.api_url = api_url.rstrip("/")
        if session is None:
            import requests
            session = requests.Session()
        self.session = session
        self._installation_ids: Dict[str, int] = {}

    @classmethod
    def create_from_env(cls, session=None) -> "GitHubApp":
        """Env contract of the reference (github_app.py:62-67):
        GITHUB_APP_ID + GITHUB_APP_PEM_KEY (path)."""
        app_id = os.environ["GITHUB_APP_ID"]
        pem_key = os.environ["GITHUB_APP_PEM_KEY"]
        return cls(pem_path=pem_key, app_id=app_id, session=session)

    def get_jwt(self) -> str:
        return app_jwt(self.app_id, self.pem, ttl_s=60)

    def _headers(self, token: str) -> dict:
        return {"Authorization": f"Bearer {token}",
                "Accept": "application/vnd.github.v3+json"}

    def get_app(self) -> dict:
        r = self.session.get(f"{self.api_url}/app",
                             headers=self._headers(self.get_jwt()))
        r.raise_for_status()
        return r.json()

    def get_installation_id(self, owner: str, repo: Optional[str] = None) -> int:
        key = f"{owner}/{repo or ''}"
        if key in self._installation_ids:
            return self._installation_ids[key]
        if repo:
            url = f"{self.api_url}/repos/{owner}/{repo}/installation"
        else:
            url = f"{self.api_url}/orgs/{owner}/installation"
        r = self.session.get(url, headers=self._headers(self.get_jwt()))
        r.raise_for_status()
        iid = r.json()["id"]
        self._installation_ids[key] = iid
        return iid

    def get_installation_access_token(self, installation_id: int) -> dict:
        r = self.session.post(
            f"{self.api_url}/app/installations/{installation_id}/access_tokens",
            headers=self._headers(self.get_jwt()))
        r.raise_for_status()
        return r.json()  # {'token': ..., 'expires_at': ISO8601}

    def list_installation_repos(self, installation_id: int) -> list:
        """Repositories an installation can access (github_app.py parity)."""
        token = self.get_installation_access_token(installation_id)["token"]
        repos, page = [], 1
        while True:
            r = self.session.get(
                f"{self.api_url}/installation/repositories",
                params={"per_page": 100, "page": page},
                headers={"Authorization": f"token {token}",
                         "Accept": "application/vnd.github.v3+json"})
            r.raise_for_status()
            batch = r.json().get("repositories", [])
            repos.extend(batch)
            if len(batch) < 100:
                return repos
            page += 1

    def add_reaction(self, owner: str, repo: str, comment_id: int,
                     content: str, token: str) -> dict:
        """React to an issue comment (worker feedback loop parity)."""
        r = self.session.post(
            f"{self.api_url}/repos/{owner}/{repo}/issues/comments/"
            f"{comment_id}/reactions",
            json={"content": content},
            headers={"Authorization": f"token {token}",
                     "Accept": "application/vnd.github.squirrel-girl-preview+json"})
        r.raise_for_status()
        return r.json()


class FixedAccessTokenGenerator:
    """A constant personal-access-token source (github_app.py:265-303)."""

    ENV_CHAIN = ("INPUT_GITHUB_PERSONAL_ACCESS_TOKEN",
                 "GITHUB_PERSONAL_ACCESS_TOKEN", "GITHUB_TOKEN")

    def __init__(self, token: str):
        self._token = token

    @property
    def token(self) -> str:
        return self._token

    @classmethod
    def from_env(cls) -> Optional["FixedAccessTokenGenerator"]:
        for name in cls.ENV_CHAIN:
            v = os.environ.get(name)
            if v:
                return cls(v)
        return None

    def auth_headers(self) -> dict:
        return {"Authorization": f"token {self.token}"}


class GitHubAppTokenGenerator:
    """Installation-token source with refresh-on-expiry (github_app.py:305-364)."""

    def __init__(self, app: GitHubApp, owner: str, repo: Optional[str] = None,
                 skew_s: int = 60):
        self.app, self.owner, self.repo, self.skew_s = app, owner, repo, skew_s
        self._token: Optional[str] = None
        self._expires: Optional[datetime.datetime] = None

    @property
    def token(self) -> str:
        now = datetime.datetime.now(datetime.timezone.utc)
        if self._token is None or self._expires is None or \
                now + datetime.timedelta(seconds=self.skew_s) >= self._expires:
            iid = self.app.get_installation_id(self.owner, self.repo)
            data = self.app.get_installation_access_token(iid)
            self._token = data["token"]
            exp = data["expires_at"].replace("Z", "+00:00")
            self._expires = datetime.datetime.fromisoformat(exp)
        return self._token

    def auth_headers(self) -> dict:
        return {"Authorization": f"token {self.token}"}
