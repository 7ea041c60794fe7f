"""Runtime command builders (reference arksapplication_controller.go:941-1014
and arksdisaggregatedapplication_controller.go:1630-1724).

The default runtime is `arks` — OUR first-party MI355X engine
(python -m arks_amd.server). vllm/sglang/dynamo command shapes are kept for
API compatibility with the reference's pluggable-runtime contract.
"""

from __future__ import annotations

import os
import shlex

DEFAULT_ARKS_IMAGE = "arks-amd/runtime:latest"
DEFAULT_VLLM_IMAGE = "vllm/vllm-openai:v0.8.2"
DEFAULT_SGLANG_IMAGE = "lmsysorg/sglang:v0.4.5-cu124"
DEFAULT_DYNAMO_IMAGE = "scitixai/k8s/dynamo:vllm"
DEFAULT_ROUTER_IMAGE = "arks-amd/router:latest"


def runtime_image(runtime: str, override: str = "") -> str:
    if override:
        return override
    env = {
        "arks": ("ARKS_RUNTIME_DEFAULT_ARKS_IMAGE", DEFAULT_ARKS_IMAGE),
        "vllm": ("ARKS_RUNTIME_DEFAULT_VLLM_IMAGE", DEFAULT_VLLM_IMAGE),
        "sglang": ("ARKS_RUNTIME_DEFAULT_SGLANG_IMAGE", DEFAULT_SGLANG_IMAGE),
        "dynamo": ("ARKS_RUNTIME_DEFAULT_DYNAMO_IMAGE", DEFAULT_DYNAMO_IMAGE),
    }[runtime]
    return os.environ.get(env[0], env[1])


def _join(args: list[str]) -> str:
    return " ".join(args)


def leader_command(runtime: str, model_path: str, served_name: str,
                   tp_size: int, common_args: list[str], group_size: int) -> list[str]:
    """The leader container command (reference :941-980)."""
    extra = _join(common_args)
    if runtime == "arks":
        cmd = (
            f"python3 -m arks_amd.server --port 8080 --model {shlex.quote(model_path)} "
            f"--served-model-name {shlex.quote(served_name)}"
        )
        if tp_size > 0:
            cmd += f" --tensor-parallel-size {tp_size}"
        if extra:
            cmd += f" {extra}"
        return ["sh", "-c", cmd]
    if runtime == "vllm":
        cmd = (
            "/vllm-workspace/examples/online_serving/multi-node-serving.sh leader "
            "--ray_cluster_size=$(LWS_GROUP_SIZE); "
            f"python3 -m vllm.entrypoints.openai.api_server --port 8080 "
            f"--model {model_path} --served-model-name {served_name}"
        )
        if tp_size > 0:
            cmd += f" --tensor-parallel-size {tp_size}"
        if extra:
            cmd += f" {extra}"
        return ["sh", "-c", cmd]
    if runtime == "sglang":
        cmd = (
            f"python3 -m sglang.launch_server --model-path {model_path} "
            f"--served-model-name {served_name} --port 8080 "
            "--dist-init-addr $(LWS_LEADER_ADDRESS):20000 "
            "--nnodes $(LWS_GROUP_SIZE) --node-rank 0 --trust-remote-code "
            "--enable-metrics"
        )
        if tp_size > 0:
            cmd += f" --tp {tp_size}"
        if extra:
            cmd += f" {extra}"
        return ["sh", "-c", cmd]
    if runtime == "dynamo":
        cmd = f"dynamo run in=http out=dyn://{served_name} {extra}".strip()
        return ["sh", "-c", cmd]
    raise ValueError(f"unsupported runtime {runtime!r}")


def worker_command(runtime: str, model_path: str, served_name: str,
                   tp_size: int, common_args: list[str]) -> list[str]:
    """The worker container command (reference :982-1014)."""
    extra = _join(common_args)
    if runtime == "arks":
        # multi-node group: workers join the leader's torch.distributed
        # rendezvous (LWS_LEADER_ADDRESS / LWS_WORKER_INDEX injected by LWS)
        cmd = (
            f"python3 -m arks_amd.server --port 8080 --model {shlex.quote(model_path)} "
            f"--served-model-name {shlex.quote(served_name)} "
            "--leader-address $(LWS_LEADER_ADDRESS) --node-rank $(LWS_WORKER_INDEX)"
        )
        if tp_size > 0:
            cmd += f" --tensor-parallel-size {tp_size}"
        if extra:
            cmd += f" {extra}"
        return ["sh", "-c", cmd]
    if runtime == "vllm":
        return [
            "sh", "-c",
            "/vllm-workspace/examples/online_serving/multi-node-serving.sh worker "
            "--ray_address=$(LWS_LEADER_ADDRESS)",
        ]
    if runtime == "sglang":
        cmd = (
            f"python3 -m sglang.launch_server --model-path {model_path} "
            "--dist-init-addr $(LWS_LEADER_ADDRESS):20000 "
            "--nnodes $(LWS_GROUP_SIZE) --node-rank $(LWS_WORKER_INDEX) "
            "--trust-remote-code"
        )
        if tp_size > 0:
            cmd += f" --tp {tp_size}"
        if extra:
            cmd += f" {extra}"
        return ["sh", "-c", cmd]
    if runtime == "dynamo":
        return ["sh", "-c", f"dynamo run in=dyn://{served_name} out=vllm {model_path}"]
    raise ValueError(f"unsupported runtime {runtime!r}")


def disagg_worker_command(role: str, model_path: str, served_name: str,
                          tp_size: int, common_args: list[str],
                          is_leader: bool) -> list[str]:
    """Prefill/decode-separated engine command (reference :1672-1724 shape,
    arks runtime)."""
    extra = _join(common_args)
    cmd = (
        f"python3 -m arks_amd.server --port 8080 --model {shlex.quote(model_path)} "
        f"--served-model-name {shlex.quote(served_name)} "
        f"--disaggregation-mode {role}"
    )
    if tp_size > 0:
        cmd += f" --tensor-parallel-size {tp_size}"
    if not is_leader:
        cmd += " --leader-address $(LWS_LEADER_ADDRESS) --node-rank $(LWS_WORKER_INDEX)"
    if extra:
        cmd += f" {extra}"
    return ["sh", "-c", cmd]


def router_command(served_name: str, namespace: str, app_name: str, port: int,
                   metric_port: int, policy: str = "cache_aware",
                   router_args: list[str] | None = None) -> list[str]:
    """PD router command — our Go-less router (arks_amd.router) with pod
    label service-discovery, mirroring sglang-router's flags
    (reference :1630-1670)."""
    cmd = (
        f"python3 -m arks_amd.router --pd-disaggregation "
        f"--service-discovery --namespace {namespace} "
        f"--prefill-selector arks.ai/application={app_name} "
        f"arks.ai/disaggregation-role=prefill arks.ai/work-load-role=leader "
        f"--decode-selector arks.ai/application={app_name} "
        f"arks.ai/disaggregation-role=decode arks.ai/work-load-role=leader "
        f"--port {port} --prometheus-port {metric_port} --policy {policy}"
    )
    if router_args:
        cmd += " " + _join(router_args)
    return ["sh", "-c", cmd]
