"""AWS account/region client context.

Parity with reference src/providers/aws/client.ts (177 LoC): cached client
factory per account/region (L78), multi-account config from services.yaml
(L127-154). There are no real SDK clients in this environment; the
"client" carries account/region context into the executor and the
registry of configured accounts.
"""
from __future__ import annotations

from dataclasses import dataclass, field
from typing import Any, Optional


@dataclass(frozen=True)
class AwsAccount:
    account_id: str
    region: str = "us-east-1"
    profile: str = "default"
    role_arn: str = ""


class AwsClientFactory:
    def __init__(self, default_region: str = "us-east-1") -> None:
        self.default_region = default_region
        self.accounts: dict[str, AwsAccount] = {}
        self._cache: dict[tuple[str, str], dict[str, Any]] = {}

    def load_accounts(self, services_config: dict[str, Any]) -> None:
        """Multi-account config from services.yaml (reference L127-154)."""
        for acct in services_config.get("aws", {}).get("accounts", []) or []:
            account = AwsAccount(
                account_id=str(acct.get("accountId", "default")),
                region=acct.get("region", self.default_region),
                profile=acct.get("profile", "default"),
                role_arn=acct.get("roleArn", ""),
            )
            self.accounts[account.account_id] = account
        if not self.accounts:
            self.accounts["default"] = AwsAccount(account_id="default", region=self.default_region)

    def get_client(self, service: str, account_id: str = "default",
                   region: Optional[str] = None) -> dict[str, Any]:
        """Cached per-(account, region) client context (reference L78)."""
        account = self.accounts.get(account_id) or AwsAccount(account_id=account_id,
                                                              region=self.default_region)
        key = (f"{account_id}:{service}", region or account.region)
        if key not in self._cache:
            self._cache[key] = {
                "service": service,
                "accountId": account.account_id,
                "region": region or account.region,
            }
        return self._cache[key]
