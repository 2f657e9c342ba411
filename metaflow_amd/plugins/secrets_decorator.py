"""@secrets: fetch secrets into the task environment at task start.

Parity target: /root/reference/metaflow/plugins/secrets/. Providers are
pluggable; the built-in 'env-file' provider reads KEY=VALUE lines from a
file named by the secret source (cloud secret managers slot in behind the
same interface when available).
"""

import os

from ..decorators import StepDecorator, make_step_decorator
from ..exceptions import MFXException


def _fetch_env_file(source):
    path = os.path.expanduser(source)
    if not os.path.exists(path):
        raise MFXException("Secret source file %s not found." % source)
    out = {}
    with open(path) as f:
        for line in f:
            line = line.strip()
            if line and not line.startswith("#") and "=" in line:
                k, _, v = line.partition("=")
                out[k.strip()] = v.strip()
    return out


SECRET_PROVIDERS = {"env-file": _fetch_env_file}


class SecretsDecorator(StepDecorator):
    name = "secrets"
    defaults = {"sources": []}

    def task_pre_step(self, step_name, task_datastore, metadata, run_id,
                      task_id, flow, graph, retry_count,
                      max_user_code_retries, ubf_context, inputs):
        sources = self.attributes.get("sources") or []
        if isinstance(sources, str):
            sources = [s for s in sources.split(";") if s]
        for source in sources:
            provider_name, _, ref = str(source).partition(":")
            provider = SECRET_PROVIDERS.get(provider_name)
            if provider is None:
                raise MFXException(
                    "Unknown secrets provider '%s' (known: %s)"
                    % (provider_name, ", ".join(SECRET_PROVIDERS)))
            for k, v in provider(ref).items():
                os.environ[k] = v


secrets = make_step_decorator(SecretsDecorator)
