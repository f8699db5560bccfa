"""Azure credentials: workload identity (federated token → AAD) and managed
identity (IMDS), with token caching.

Spec: reference pkg/auth/cred.go:49-135 — self-hosted mode exchanges the
projected federated token file for an AAD access token via the OAuth2
client-credentials grant with client_assertion (what MSAL confidential
client does under the hood; here the flow is implemented directly over
httpx), caching the token-file read for 5 min and the access token until
expiry. Managed mode queries the IMDS endpoint like DefaultAzureCredential
(reference azure_client.go:78-89).
"""
from __future__ import annotations

import asyncio
import logging
import os
import time
from dataclasses import dataclass
from typing import Optional

import httpx

from .config import AzureConfig, DEPLOYMENT_MODE_MANAGED

log = logging.getLogger(__name__)

ARM_SCOPE = "https://management.azure.com/.default"
IMDS_TOKEN_URL = "http://169.254.169.254/metadata/identity/oauth2/token"
# RFC 7523 §2.2: the CLIENT-ASSERTION-TYPE urn (NOT the jwt-bearer GRANT
# urn, which AAD rejects with AADSTS50027 in this parameter)
CLIENT_ASSERTION_TYPE = "urn:ietf:params:oauth:client-assertion-type:jwt-bearer"
TOKEN_FILE_CACHE_SECONDS = 300.0  # reference cred.go:125-135
TOKEN_REFRESH_SKEW = 120.0


class CredentialError(RuntimeError):
    pass


@dataclass
class AccessToken:
    token: str
    expires_at: float  # unix seconds

    @property
    def valid(self) -> bool:
        return bool(self.token) and time.time() < self.expires_at - TOKEN_REFRESH_SKEW


class TokenCredential:
    """Interface: get_token() -> str (cached, auto-refreshing)."""

    async def get_token(self) -> str:
        raise NotImplementedError


class WorkloadIdentityCredential(TokenCredential):
    """Federated-token-file JWT → AAD access token (self-hosted mode)."""

    def __init__(self, config: AzureConfig, http: Optional[httpx.AsyncClient] = None):
        self.config = config
        self.http = http or httpx.AsyncClient(timeout=30.0)
        self._assertion: str = ""
        self._assertion_read_at: float = 0.0
        self._token: Optional[AccessToken] = None
        self._lock = asyncio.Lock()

    def _read_assertion(self) -> str:
        now = time.time()
        if self._assertion and now - self._assertion_read_at < TOKEN_FILE_CACHE_SECONDS:
            return self._assertion
        path = self.config.federated_token_file
        if not os.path.exists(path):
            raise CredentialError(
                f"federated token file {path} not found — is the workload identity "
                "webhook configured? (azure.workload.identity/use: 'true' on the pod, "
                "federated credential on the managed identity)"
            )
        with open(path) as f:
            self._assertion = f.read().strip()
        self._assertion_read_at = now
        return self._assertion

    async def get_token(self) -> str:
        async with self._lock:
            if self._token is not None and self._token.valid:
                return self._token.token
            assertion = self._read_assertion()
            url = (
                f"{self.config.authority_host.rstrip('/')}/{self.config.tenant_id}"
                "/oauth2/v2.0/token"
            )
            resp = await self.http.post(
                url,
                data={
                    "grant_type": "client_credentials",
                    "client_id": self.config.client_id,
                    "scope": ARM_SCOPE,
                    "client_assertion": assertion,
                    "client_assertion_type": CLIENT_ASSERTION_TYPE,
                },
            )
            if resp.status_code != 200:
                raise CredentialError(
                    f"AAD token exchange failed ({resp.status_code}): {resp.text[:300]}"
                )
            body = resp.json()
            self._token = AccessToken(
                token=body["access_token"],
                expires_at=time.time() + float(body.get("expires_in", 3600)),
            )
            return self._token.token


class ManagedIdentityCredential(TokenCredential):
    """IMDS-issued token (managed mode)."""

    def __init__(self, config: AzureConfig, http: Optional[httpx.AsyncClient] = None):
        self.config = config
        self.http = http or httpx.AsyncClient(timeout=10.0)
        self._token: Optional[AccessToken] = None
        self._lock = asyncio.Lock()

    async def get_token(self) -> str:
        async with self._lock:
            if self._token is not None and self._token.valid:
                return self._token.token
            params = {
                "api-version": "2018-02-01",
                "resource": "https://management.azure.com/",
            }
            if self.config.client_id:
                params["client_id"] = self.config.client_id
            resp = await self.http.get(
                IMDS_TOKEN_URL, params=params, headers={"Metadata": "true"}
            )
            if resp.status_code != 200:
                raise CredentialError(
                    f"IMDS token request failed ({resp.status_code}): {resp.text[:300]}"
                )
            body = resp.json()
            self._token = AccessToken(
                token=body["access_token"],
                expires_at=float(body.get("expires_on", time.time() + 3600)),
            )
            return self._token.token


class StaticCredential(TokenCredential):
    """Fixed token — tests and local development."""

    def __init__(self, token: str = "test-token"):
        self._token = token

    async def get_token(self) -> str:
        return self._token


def new_credential(config: AzureConfig, http: Optional[httpx.AsyncClient] = None) -> TokenCredential:
    """Credential per deployment mode (reference azure_client.go:74-111)."""
    if config.deployment_mode == DEPLOYMENT_MODE_MANAGED:
        return ManagedIdentityCredential(config, http)
    return WorkloadIdentityCredential(config, http)
