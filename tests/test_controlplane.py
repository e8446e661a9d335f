"""Control-plane tests.

Tier 1 mirrors the reference's table-driven builder tests (SURVEY.md §4.1:
lws_test / podgroup_test / epp_test / strategy_test / inferencepool_test /
httproute_test / hash_test); tier 2 mirrors its envtest integration suite
(§4.2) against the in-memory apiserver, including the metadata-only-change
no-op invariant and the readiness aggregation the reference left untested.
"""

import yaml

from fusioninfer_amd.controlplane import api, router, scheduling, workload
from fusioninfer_amd.controlplane.fake import FakeClient
from fusioninfer_amd.controlplane.hashutil import compute_spec_hash
from fusioninfer_amd.controlplane.reconciler import InferenceServiceReconciler


def pod_template(image="fusioninfer-amd/engine:latest", gpus=1, args=None):
    return {
        "spec": {
            "containers": [
                {
                    "name": "engine",
                    "image": image,
                    "command": ["python", "-m", "fusioninfer_amd.server"],
                    "args": args or ["--model", "Qwen3-8B"],
                    "resources": {"limits": {"amd.com/gpu": gpus}},
                }
            ]
        }
    }


def monolithic_svc(name="svc", replicas=1, node_count=1):
    role = api.Role(
        component_type=api.WORKER,
        replicas=replicas,
        template=pod_template(),
        multinode=api.Multinode(node_count) if node_count > 1 else None,
    )
    return api.InferenceService(name=name, roles=[role], uid="uid-1")


def pd_svc(name="pd"):
    return api.InferenceService(
        name=name,
        uid="uid-2",
        roles=[
            api.Role(api.PREFILLER, 1, pod_template()),
            api.Role(api.DECODER, 1, pod_template()),
            api.Role(
                api.ROUTER, 1, routing_strategy=api.PD_DISAGGREGATION,
                httproute={"parentRefs": [{"name": "gw"}]},
            ),
        ],
    )


# ---------------------------------------------------------------- workload
class TestLWSBuilder:
    def test_naming(self):
        svc = monolithic_svc("m")
        assert workload.generate_lws_name(svc.name, svc.roles[0], 2) == "m-worker-2"

    def test_labels_and_per_replica_mode(self):
        svc = monolithic_svc()
        lws = workload.build_lws(svc, svc.roles[0], 0)
        labels = lws["metadata"]["labels"]
        assert labels[workload.LABEL_SERVICE] == "svc"
        assert labels[workload.LABEL_ROLE] == "worker"
        assert labels[workload.LABEL_REPLICA_INDEX] == "0"
        assert labels[workload.LABEL_COMPONENT_TYPE] == "worker"
        assert labels[workload.LABEL_MANAGED_BY] == workload.MANAGED_BY
        assert workload.LABEL_SPEC_HASH in labels
        assert lws["spec"]["replicas"] == 1
        assert lws["spec"]["leaderWorkerTemplate"]["size"] == 1
        assert lws["spec"]["startupPolicy"] == "LeaderCreated"

    def test_is_multi_node(self):
        assert not workload.is_multi_node(api.Role(multinode=None))
        assert not workload.is_multi_node(api.Role(multinode=api.Multinode(1)))
        assert workload.is_multi_node(api.Role(multinode=api.Multinode(2)))

    def test_multinode_command_wrapping(self):
        svc = monolithic_svc(node_count=2)
        lws = workload.build_lws(svc, svc.roles[0], 0)
        lwt = lws["spec"]["leaderWorkerTemplate"]
        assert lwt["size"] == 2
        leader = lwt["leaderTemplate"]["spec"]["containers"][0]
        worker = lwt["workerTemplate"]["spec"]["containers"][0]
        assert leader["command"] == ["sh", "-c"]
        assert "--node-rank 0" in leader["args"][0]
        assert "--nnodes 2" in leader["args"][0]
        assert "$LWS_LEADER_ADDRESS" in leader["args"][0]
        assert "--master-port 29500" in leader["args"][0]
        # torchrun rendezvous, never Ray
        assert "ray" not in leader["args"][0]
        assert "--node-rank $LWS_WORKER_INDEX" in worker["args"][0]
        # leader readiness probes the rendezvous port
        assert leader["readinessProbe"]["tcpSocket"]["port"] == 29500

    def test_gang_annotations_and_scheduler(self):
        svc = monolithic_svc(node_count=2)
        lws = workload.build_lws(
            svc, svc.roles[0], 0, gang_scheduled=True,
            podgroup_name="svc", task_name="worker-0",
        )
        meta = lws["spec"]["leaderWorkerTemplate"]["leaderTemplate"]["metadata"]
        assert meta["annotations"][workload.ANNOTATION_PODGROUP] == "svc"
        assert meta["annotations"][workload.ANNOTATION_TASK_SPEC] == "worker-0"
        pod_spec = lws["spec"]["leaderWorkerTemplate"]["leaderTemplate"]["spec"]
        assert pod_spec["schedulerName"] == "volcano"


# -------------------------------------------------------------- scheduling
class TestGangScheduling:
    def test_pd_detection(self):
        assert scheduling.is_pd_disaggregated(pd_svc())
        assert not scheduling.is_pd_disaggregated(monolithic_svc())

    def test_gang_policy(self):
        assert not scheduling.needs_gang_scheduling(monolithic_svc())
        assert scheduling.needs_gang_scheduling(monolithic_svc(node_count=2))
        assert scheduling.needs_gang_scheduling(pd_svc())

    def test_podgroup_members(self):
        svc = monolithic_svc(replicas=2, node_count=3)
        pg = scheduling.build_podgroup(svc)
        assert pg["spec"]["minMember"] == 6
        assert pg["spec"]["minTaskMember"] == {"worker-0": 3, "worker-1": 3}
        # 2 replicas x 3 nodes x 1 gpu
        assert pg["spec"]["minResources"]["amd.com/gpu"] == 6

    def test_podgroup_pd(self):
        pg = scheduling.build_podgroup(pd_svc())
        assert pg["spec"]["minMember"] == 2
        assert pg["spec"]["minTaskMember"] == {"prefiller-0": 1, "decoder-0": 1}
        assert pg["metadata"]["name"] == "pd"

    def test_counts(self):
        svc = monolithic_svc(replicas=2, node_count=3)
        assert scheduling.get_replica_count(svc) == 2
        assert scheduling.get_node_count(svc) == 6


# ------------------------------------------------------------------ router
class TestStrategies:
    def test_custom_config_passthrough(self):
        svc = monolithic_svc()
        role = api.Role(api.ROUTER, endpoint_picker_config="my: config\n")
        assert router.generate_epp_config(svc, role) == "my: config\n"

    def test_default_is_prefix_cache(self):
        svc = monolithic_svc()
        cfg = yaml.safe_load(router.generate_epp_config(svc, api.Role(api.ROUTER)))
        types = [p["type"] for p in cfg["plugins"]]
        assert "prefix-cache-scorer" in types and "max-score-picker" in types
        params = cfg["plugins"][0]["parameters"]
        assert params == {
            "blockSize": 5,
            "maxPrefixBlocksToMatch": 256,
            "lruCapacityPerServer": 31250,
        }
        prof = cfg["schedulingProfiles"][0]
        assert prof["name"] == "default"
        assert {"pluginRef": "prefix-cache-scorer", "weight": 100} in prof["plugins"]

    def test_each_simple_strategy(self):
        svc = monolithic_svc()
        for strategy, scorer in [
            (api.KV_CACHE_UTILIZATION, "kv-cache-utilization-scorer"),
            (api.QUEUE_SIZE, "queue-scorer"),
            (api.LORA_AFFINITY, "lora-affinity-scorer"),
        ]:
            cfg = yaml.safe_load(
                router.generate_epp_config(
                    svc, api.Role(api.ROUTER, routing_strategy=strategy)
                )
            )
            assert {"type": scorer} in cfg["plugins"]

    def test_pd_strategy(self):
        svc = pd_svc()
        cfg = yaml.safe_load(
            router.generate_epp_config(svc, svc.router_roles()[0])
        )
        types = [p["type"] for p in cfg["plugins"]]
        assert "pd-profile-handler" in types
        assert "prefill-header-handler" in types
        assert types.count("by-label") == 2
        by_label = [p for p in cfg["plugins"] if p["type"] == "by-label"]
        assert by_label[0]["parameters"]["validValues"] == ["prefiller"]
        assert by_label[1]["parameters"]["validValues"] == ["decoder"]
        profiles = {p["name"] for p in cfg["schedulingProfiles"]}
        assert profiles == {"prefill", "decode"}
        handler = cfg["plugins"][0]["parameters"]
        assert handler == {"threshold": 0, "hashBlockSize": 5, "primaryPort": 8000}

    def test_pd_falls_back_when_not_pd(self):
        svc = monolithic_svc()
        cfg = yaml.safe_load(
            router.generate_epp_config(
                svc, api.Role(api.ROUTER, routing_strategy=api.PD_DISAGGREGATION)
            )
        )
        assert cfg["plugins"][0]["type"] == "prefix-cache-scorer"


class TestEPPBuilders:
    def test_deployment(self):
        svc = monolithic_svc()
        dep = router.build_epp_deployment(svc)
        c = dep["spec"]["template"]["spec"]["containers"][0]
        assert f"--pool-name={router.pool_name(svc)}" in c["args"]
        ports = {p["containerPort"] for p in c["ports"]}
        assert ports == {9002, 9003, 9090}
        assert dep["spec"]["strategy"]["type"] == "Recreate"
        assert dep["spec"]["replicas"] == 1
        assert c["livenessProbe"]["grpc"]["port"] == 9003

    def test_image_env_override(self, monkeypatch):
        monkeypatch.setenv("EPP_IMAGE", "my-registry/epp:v9")
        assert router.get_epp_image() == "my-registry/epp:v9"
        monkeypatch.delenv("EPP_IMAGE")
        assert router.get_epp_image() == router.DEFAULT_EPP_IMAGE

    def test_service_ports(self):
        s = router.build_epp_service(monolithic_svc())
        assert {p["port"] for p in s["spec"]["ports"]} == {9002, 9003, 9090}

    def test_rbac(self):
        r = router.build_epp_role(monolithic_svc())
        resources = {res for rule in r["rules"] for res in rule["resources"]}
        assert {"pods", "inferencepools", "leases", "events"} <= resources


class TestInferencePool:
    def test_selector_single_worker_role(self):
        svc = monolithic_svc()
        sel = router.build_pool_selector(svc)
        assert sel[workload.LABEL_SERVICE] == "svc"
        assert sel[workload.LABEL_COMPONENT_TYPE] == "worker"
        # only leader pods are routable
        assert sel[workload.LABEL_LWS_WORKER_INDEX] == "0"

    def test_selector_many_worker_roles(self):
        svc = pd_svc()
        sel = router.build_pool_selector(svc)
        assert workload.LABEL_COMPONENT_TYPE not in sel
        assert sel[workload.LABEL_LWS_WORKER_INDEX] == "0"

    def test_pool_spec(self):
        svc = monolithic_svc()
        pool = router.build_inference_pool(svc)
        assert pool["spec"]["targetPorts"] == [{"number": 8000}]
        epr = pool["spec"]["endpointPickerRef"]
        assert epr["name"] == "svc-epp" and epr["port"]["number"] == 9002


class TestHTTPRoute:
    def test_user_spec_merged_rules_overridden(self):
        svc = pd_svc()
        role = svc.router_roles()[0]
        role.httproute = {
            "parentRefs": [{"name": "gw", "sectionName": "http"}],
            "hostnames": ["x.example.com"],
            "rules": [{"backendRefs": [{"name": "user-backend"}]}],
        }
        route = router.build_httproute(svc, role)
        assert route["spec"]["parentRefs"][0]["sectionName"] == "http"
        assert route["spec"]["hostnames"] == ["x.example.com"]
        refs = route["spec"]["rules"][0]["backendRefs"]
        assert refs == [
            {
                "group": "inference.networking.k8s.io",
                "kind": "InferencePool",
                "name": "pd-pool",
            }
        ]


# -------------------------------------------------------------------- hash
class TestSpecHash:
    def test_deterministic(self):
        svc = monolithic_svc()
        a = workload.build_lws(svc, svc.roles[0], 0)
        b = workload.build_lws(svc, svc.roles[0], 0)
        assert (
            a["metadata"]["labels"][workload.LABEL_SPEC_HASH]
            == b["metadata"]["labels"][workload.LABEL_SPEC_HASH]
        )

    def test_sensitive_to_spec_changes(self):
        svc = monolithic_svc()
        a = workload.build_lws(svc, svc.roles[0], 0)
        svc.roles[0].template["spec"]["containers"][0]["image"] = "other:img"
        b = workload.build_lws(svc, svc.roles[0], 0)
        assert (
            a["metadata"]["labels"][workload.LABEL_SPEC_HASH]
            != b["metadata"]["labels"][workload.LABEL_SPEC_HASH]
        )

    def test_key_order_invariant(self):
        assert compute_spec_hash({"a": 1, "b": 2}) == compute_spec_hash(
            {"b": 2, "a": 1}
        )


# ------------------------------------------------------- reconcile (tier 2)
class TestReconcile:
    def setup_method(self):
        self.client = FakeClient()
        self.rec = InferenceServiceReconciler(self.client)

    def _apply(self, svc: api.InferenceService):
        existing = self.client.try_get("InferenceService", svc.name, svc.namespace)
        if existing is None:
            self.client.create(svc.to_dict())
        else:
            d = svc.to_dict()
            d["metadata"]["generation"] = (
                existing["metadata"].get("generation", 1) + 1
            )
            self.client.update(d)
        self.rec.reconcile(svc.name, svc.namespace)

    def test_lws_created_on_cr_create(self):
        self._apply(monolithic_svc())
        assert self.client.try_get("LeaderWorkerSet", "svc-worker-0") is not None

    def test_replica_increase_creates_new_lws(self):
        svc = monolithic_svc(replicas=1)
        self._apply(svc)
        svc.roles[0].replicas = 3
        self._apply(svc)
        names = {
            o["metadata"]["name"]
            for o in self.client.list("LeaderWorkerSet")
        }
        assert names == {"svc-worker-0", "svc-worker-1", "svc-worker-2"}

    def test_replica_decrease_cleans_orphans(self):
        svc = monolithic_svc(replicas=3)
        self._apply(svc)
        svc.roles[0].replicas = 1
        self._apply(svc)
        names = {
            o["metadata"]["name"] for o in self.client.list("LeaderWorkerSet")
        }
        assert names == {"svc-worker-0"}

    def test_metadata_only_change_is_noop(self):
        svc = monolithic_svc()
        self._apply(svc)
        rv_before = self.client.get("LeaderWorkerSet", "svc-worker-0")[
            "metadata"
        ]["resourceVersion"]
        svc.labels["new-label"] = "x"
        self._apply(svc)
        rv_after = self.client.get("LeaderWorkerSet", "svc-worker-0")[
            "metadata"
        ]["resourceVersion"]
        assert rv_before == rv_after

    def test_image_change_propagates(self):
        svc = monolithic_svc()
        self._apply(svc)
        before = self.client.get("LeaderWorkerSet", "svc-worker-0")
        svc.roles[0].template["spec"]["containers"][0]["image"] = "new:img"
        self._apply(svc)
        after = self.client.get("LeaderWorkerSet", "svc-worker-0")
        assert (
            before["metadata"]["resourceVersion"]
            != after["metadata"]["resourceVersion"]
        )
        img = after["spec"]["leaderWorkerTemplate"]["leaderTemplate"]["spec"][
            "containers"
        ][0]["image"]
        assert img == "new:img"

    def test_args_change_propagates(self):
        svc = monolithic_svc()
        self._apply(svc)
        svc.roles[0].template["spec"]["containers"][0]["args"] = [
            "--model", "Llama-3-70B",
        ]
        before_rv = self.client.get("LeaderWorkerSet", "svc-worker-0")[
            "metadata"
        ]["resourceVersion"]
        self._apply(svc)
        after = self.client.get("LeaderWorkerSet", "svc-worker-0")
        assert after["metadata"]["resourceVersion"] != before_rv

    def test_pd_service_full_stack(self):
        svc = pd_svc()
        self._apply(svc)
        assert self.client.try_get("PodGroup", "pd") is not None
        assert self.client.try_get("LeaderWorkerSet", "pd-prefiller-0") is not None
        assert self.client.try_get("LeaderWorkerSet", "pd-decoder-0") is not None
        assert self.client.try_get("Deployment", "pd-epp") is not None
        assert self.client.try_get("Service", "pd-epp") is not None
        assert self.client.try_get("ConfigMap", "pd-epp-config") is not None
        assert self.client.try_get("ServiceAccount", "pd-epp") is not None
        assert self.client.try_get("Role", "pd-epp") is not None
        assert self.client.try_get("RoleBinding", "pd-epp") is not None
        assert self.client.try_get("InferencePool", "pd-pool") is not None
        assert self.client.try_get("HTTPRoute", "pd-route") is not None
        cm = self.client.get("ConfigMap", "pd-epp-config")
        assert "pd-profile-handler" in cm["data"]["config.yaml"]

    def test_no_podgroup_for_simple_service(self):
        self._apply(monolithic_svc())
        assert self.client.try_get("PodGroup", "svc") is None

    def test_status_pending_then_active(self):
        svc = monolithic_svc()
        self._apply(svc)
        st = self.client.get("InferenceService", "svc")["status"]
        comp = st["components"]["worker"]
        assert comp["phase"] == "Deploying"  # LWS exists, not ready
        conds = {c["type"]: c["status"] for c in st["conditions"]}
        assert conds["Initialized"] == "True"
        assert conds["Active"] == "False"
        # fake LWS readiness (the reference's envtest gap)
        self.client.set_lws_ready("svc-worker-0")
        self.rec.reconcile("svc")
        st = self.client.get("InferenceService", "svc")["status"]
        assert st["components"]["worker"]["phase"] == "Running"
        conds = {c["type"]: c["status"] for c in st["conditions"]}
        assert conds["Active"] == "True"
        assert conds["Failed"] == "False"

    def test_multinode_status_counts_pods(self):
        svc = monolithic_svc(replicas=2, node_count=3)
        self._apply(svc)
        self.client.set_lws_ready("svc-worker-0")
        self.rec.reconcile("svc")
        comp = self.client.get("InferenceService", "svc")["status"][
            "components"
        ]["worker"]
        assert comp["readyReplicas"] == 1
        assert comp["readyPods"] == 3
        assert comp["phase"] == "Deploying"


class TestCLIAndCRD:
    def test_crd_renders(self):
        from fusioninfer_amd.controlplane.crd import inference_service_crd

        crd = inference_service_crd()
        assert crd["metadata"]["name"] == "inferenceservices.fusioninfer.io"
        v = crd["spec"]["versions"][0]
        role = v["schema"]["openAPIV3Schema"]["properties"]["spec"][
            "properties"]["roles"]["items"]
        assert set(role["properties"]["componentType"]["enum"]) == {
            "router", "prefiller", "decoder", "worker"
        }
        assert "status" in v["subresources"]

    def test_render_cli_on_samples(self, tmp_path, capsys):
        import glob

        from fusioninfer_amd.controlplane.__main__ import main

        for sample in sorted(glob.glob("config/samples/*.yaml")):
            assert main(["render", sample]) == 0
            out = capsys.readouterr().out
            docs = [d for d in yaml.safe_load_all(out) if d]
            kinds = {d["kind"] for d in docs}
            assert "LeaderWorkerSet" in kinds
            assert "InferencePool" in kinds
            assert "HTTPRoute" in kinds
            # every worker pod asks for amd.com/gpu, never nvidia.com/gpu
            assert "nvidia.com/gpu" not in out
            if "pd" in sample or "tp8" in sample:
                assert "PodGroup" in kinds


class TestModelLoaderStub:
    def test_roundtrip_and_noop_reconcile(self):
        from fusioninfer_amd.controlplane.api import ModelLoader
        from fusioninfer_amd.controlplane.reconciler import ModelLoaderReconciler

        client = FakeClient()
        ml = ModelLoader("demo", foo="bar")
        client.create(ml.to_dict())
        rec = ModelLoaderReconciler(client)
        assert rec.reconcile("demo") is None
        back = ModelLoader.from_dict(client.get("ModelLoader", "demo"))
        assert back.foo == "bar"
