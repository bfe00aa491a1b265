"""W3C DID identities + Verifiable Credentials for the execution audit ledger.

Reimplements the reference's DID/VC capability natively:
Ed25519 signing runs in the C++ extension (libcrypto); key derivation is the
same simplified scheme (child seed = SHA-256(parent_seed || path)); DID
encoding is did:key with multicodec 0xed01 + base64url (matching the
reference's deliberate base64 choice, did_service.go:528-538).

Hierarchy: server root DID -> agent DIDs (one per node, derivation index)
-> component DIDs (one per reasoner/skill).
"""
from __future__ import annotations

import base64
import hashlib
import json
import os
import secrets
import time
from pathlib import Path

try:
    from agentfield_amd import _native as _crypto
except ImportError:  # pragma: no cover - native ext should normally be built
    _crypto = None

ED25519_MULTICODEC = b"\xed\x01"


def _b64u(b: bytes) -> str:
    return base64.urlsafe_b64encode(b).decode().rstrip("=")


def _b64u_dec(s: str) -> bytes:
    return base64.urlsafe_b64decode(s + "=" * (-len(s) % 4))


def canonical_json(obj) -> bytes:
    return json.dumps(obj, sort_keys=True, separators=(",", ":"),
                      ensure_ascii=False).encode()


def sha256_hex(data: bytes) -> str:
    return hashlib.sha256(data).hexdigest()


class Keystore:
    """File keystore holding the master seed (0600 perms).  When
    AGENTFIELD_KEYSTORE_KEY (base64url, 32 bytes) is set the seed is
    AES-256-GCM encrypted at rest via the native libcrypto extension
    (reference parity: C23 keystore encryption)."""

    def __init__(self, path: str | None, kek: bytes | None = None):
        self.path = Path(path) if path else None
        env = os.environ.get("AGENTFIELD_KEYSTORE_KEY")
        self.kek = kek or (_b64u_dec(env) if env else None)
        self._seed: bytes | None = None

    def _encode(self, seed: bytes) -> str:
        if self.kek:
            return "enc:" + _b64u(bytes(_crypto.aes_gcm_encrypt(self.kek, seed)))
        return _b64u(seed)

    def _decode(self, text: str) -> bytes:
        text = text.strip()
        if text.startswith("enc:"):
            if not self.kek:
                raise ValueError("keystore is encrypted; set "
                                 "AGENTFIELD_KEYSTORE_KEY")
            return bytes(_crypto.aes_gcm_decrypt(self.kek,
                                                 _b64u_dec(text[4:])))
        return _b64u_dec(text)

    @property
    def seed(self) -> bytes:
        if self._seed is None:
            if self.path and self.path.exists():
                self._seed = self._decode(self.path.read_text())
            else:
                self._seed = secrets.token_bytes(32)
                if self.path:
                    self.path.parent.mkdir(parents=True, exist_ok=True)
                    self.path.write_text(self._encode(self._seed))
                    os.chmod(self.path, 0o600)
        return self._seed


def derive_seed(parent: bytes, path: str) -> bytes:
    return hashlib.sha256(parent + path.encode()).digest()


def pubkey(seed: bytes) -> bytes:
    if _crypto is not None:
        return bytes(_crypto.ed25519_pubkey(seed))
    raise RuntimeError("native crypto extension not built")


def sign(seed: bytes, msg: bytes) -> bytes:
    return bytes(_crypto.ed25519_sign(seed, msg))


def verify(pub: bytes, msg: bytes, sig: bytes) -> bool:
    return bool(_crypto.ed25519_verify(pub, msg, sig))


def did_from_pubkey(pub: bytes) -> str:
    return "did:key:z" + _b64u(ED25519_MULTICODEC + pub)


def pubkey_from_did(did: str) -> bytes | None:
    if not did.startswith("did:key:z"):
        return None
    try:
        raw = _b64u_dec(did[len("did:key:z"):])
    except Exception:
        return None
    if not raw.startswith(ED25519_MULTICODEC):
        return None
    return raw[2:]


def did_document(did: str, pub: bytes) -> dict:
    vm = {
        "id": f"{did}#key-1",
        "type": "Ed25519VerificationKey2020",
        "controller": did,
        "publicKeyJwk": {"kty": "OKP", "crv": "Ed25519", "x": _b64u(pub)},
    }
    return {
        "@context": ["https://www.w3.org/ns/did/v1"],
        "id": did,
        "verificationMethod": [vm],
        "authentication": [vm["id"]],
        "assertionMethod": [vm["id"]],
    }


class DIDService:
    def __init__(self, storage, keystore: Keystore):
        self.storage = storage
        self.keystore = keystore
        self._seeds: dict[str, bytes] = {}  # did -> seed (in-memory only)
        self._ensure_root()

    def _register(self, seed: bytes, kind: str, subject_id: str,
                  parent_did: str | None, index: int) -> str:
        pub = pubkey(seed)
        did = did_from_pubkey(pub)
        self._seeds[did] = seed
        self.storage.put_did({
            "did": did, "kind": kind, "subject_id": subject_id,
            "parent_did": parent_did, "public_key_b64": _b64u(pub),
            "document": did_document(did, pub), "derivation_index": index,
        })
        return did

    def _ensure_root(self) -> None:
        existing = self.storage.did_for_subject("server", "root")
        if existing:
            self.root_did = existing["did"]
            self._seeds[self.root_did] = self.keystore.seed
            return
        self.root_did = self._register(self.keystore.seed, "server", "root",
                                       None, 0)

    def agent_did(self, node_id: str) -> str:
        existing = self.storage.did_for_subject("agent", node_id)
        if existing:
            did = existing["did"]
            if did not in self._seeds:
                self._seeds[did] = derive_seed(
                    self.keystore.seed, f"agent/{node_id}")
            return did
        idx = self.storage.max_derivation_index() + 1
        seed = derive_seed(self.keystore.seed, f"agent/{node_id}")
        return self._register(seed, "agent", node_id, self.root_did, idx)

    def component_did(self, node_id: str, component: str) -> str:
        subject = f"{node_id}.{component}"
        existing = self.storage.did_for_subject("component", subject)
        if existing:
            did = existing["did"]
            if did not in self._seeds:
                self._seeds[did] = derive_seed(
                    self.keystore.seed, f"component/{subject}")
            return did
        agent = self.agent_did(node_id)
        seed = derive_seed(self.keystore.seed, f"component/{subject}")
        idx = self.storage.max_derivation_index() + 1
        return self._register(seed, "component", subject, agent, idx)

    def register_node(self, node_id: str, reasoners: list[str],
                      skills: list[str]) -> dict:
        """Differential (re-)registration: mint agent + component DIDs."""
        out = {
            "agent_did": self.agent_did(node_id),
            "reasoner_dids": {r: self.component_did(node_id, r) for r in reasoners},
            "skill_dids": {s: self.component_did(node_id, s) for s in skills},
        }
        return out

    def resolve(self, did: str) -> dict | None:
        rec = self.storage.get_did(did)
        return rec["document"] if rec else None

    def seed_for(self, did: str) -> bytes | None:
        return self._seeds.get(did)


class VCService:
    """Execution Verifiable Credentials."""

    CONTEXT = ["https://www.w3.org/2018/credentials/v1",
               "https://agentfield.local/contexts/execution/v1"]

    def __init__(self, storage, dids: DIDService):
        self.storage = storage
        self.dids = dids

    def issue_execution_vc(self, execution: dict, caller_did: str | None = None,
                           target_did: str | None = None) -> dict:
        issuer = caller_did or self.dids.root_did
        vc_id = f"urn:agentfield:vc:{execution['id']}"
        input_hash = sha256_hex(canonical_json(execution.get("input")))
        output_hash = sha256_hex(canonical_json(execution.get("result")))
        doc = {
            "@context": self.CONTEXT,
            "type": ["VerifiableCredential", "AgentFieldExecutionCredential"],
            "id": vc_id,
            "issuer": issuer,
            "issuanceDate": time.strftime("%Y-%m-%dT%H:%M:%SZ", time.gmtime()),
            "credentialSubject": {
                "execution_id": execution["id"],
                "workflow_id": execution.get("run_id"),
                "session_id": execution.get("session_id"),
                "caller": {"did": issuer, "type": "agent"},
                "target": {
                    "did": target_did,
                    "agent_node_did": target_did,
                    "function_name": execution.get("reasoner_id"),
                },
                "execution": {
                    "input_hash": input_hash,
                    "output_hash": output_hash,
                    "timestamp": execution.get("finished_at"),
                    "duration_ms": execution.get("duration_ms"),
                    "status": execution.get("status"),
                    "error_message": execution.get("error_message"),
                },
                "audit": {"input_data_hash": input_hash,
                          "output_data_hash": output_hash},
            },
        }
        seed = self.dids.seed_for(issuer)
        if seed is None:
            raise ValueError(f"no signing key for issuer {issuer}")
        sig = sign(seed, canonical_json(doc))
        doc["proof"] = {
            "type": "Ed25519Signature2020",
            "created": doc["issuanceDate"],
            "verificationMethod": f"{issuer}#key-1",
            "proofPurpose": "assertionMethod",
            "proofValue": _b64u(sig),
        }
        self.storage.put_vc(vc_id, execution["id"], execution.get("run_id"),
                            issuer, doc)
        return doc

    @staticmethod
    def verify_document(doc: dict) -> dict:
        """Offline verification (also used by `af vc verify`)."""
        result = {"valid": False, "checks": {}}
        proof = doc.get("proof")
        if not proof:
            result["error"] = "missing proof"
            return result
        issuer = doc.get("issuer", "")
        pub = pubkey_from_did(issuer)
        result["checks"]["issuer_did_parses"] = pub is not None
        if pub is None:
            return result
        unsigned = {k: v for k, v in doc.items() if k != "proof"}
        try:
            sig = _b64u_dec(proof.get("proofValue", ""))
        except Exception:
            result["checks"]["proof_encoding"] = False
            return result
        ok = verify(pub, canonical_json(unsigned), sig)
        result["checks"]["signature"] = ok
        result["checks"]["type"] = "VerifiableCredential" in doc.get("type", [])
        result["valid"] = ok and result["checks"]["type"]
        return result

    def verify_execution(self, execution_id: str) -> dict:
        rec = self.storage.vc_for_execution(execution_id)
        if not rec:
            return {"valid": False, "error": "no VC for execution"}
        res = self.verify_document(rec["document"])
        exec_rec = self.storage.get_execution(execution_id)
        if exec_rec and res["valid"]:
            want = sha256_hex(canonical_json(exec_rec.get("result")))
            got = rec["document"]["credentialSubject"]["execution"]["output_hash"]
            res["checks"]["output_hash_matches"] = want == got
            res["valid"] = res["valid"] and want == got
        return res

    # -------------------------------------------------- comprehensive
    # Reference parity: VerifyExecutionVCComprehensive
    # (vc_service.go:926-1400) — integrity / security / compliance
    # sections, typed issues, 0-100 scoring
    # (score = 100 - 25*critical - 5*warning, averaged with the security
    # score), served at POST /api/ui/v1/executions/:id/verify-vc.
    @staticmethod
    def _issue(type_, severity, description, component="", field="",
               expected="", actual=""):
        return {"type": type_, "severity": severity, "component": component,
                "field": field, "expected": str(expected),
                "actual": str(actual), "description": description}

    def _integrity_checks(self, rec: dict, doc: dict,
                          exec_rec: dict | None) -> dict:
        from . import status as st
        issues = []
        subj = doc.get("credentialSubject", {})
        ex = subj.get("execution", {})
        if rec.get("issuer_did") and doc.get("issuer") != rec["issuer_did"]:
            issues.append(self._issue(
                "issuer_mismatch", "critical", "stored issuer differs from "
                "document issuer", field="issuer",
                expected=rec["issuer_did"], actual=doc.get("issuer")))
        if subj.get("execution_id") != rec.get("execution_id"):
            issues.append(self._issue(
                "execution_id_mismatch", "critical",
                "credentialSubject.execution_id differs from record",
                field="execution_id", expected=rec.get("execution_id"),
                actual=subj.get("execution_id")))
        if rec.get("run_id") and subj.get("workflow_id") != rec.get("run_id"):
            issues.append(self._issue(
                "workflow_id_mismatch", "warning",
                "workflow id differs from record", field="workflow_id",
                expected=rec.get("run_id"), actual=subj.get("workflow_id")))
        hash_ok = True
        ts_ok = True
        if exec_rec is not None:
            if subj.get("session_id") != exec_rec.get("session_id"):
                issues.append(self._issue(
                    "session_id_mismatch", "warning",
                    "session id differs from execution record",
                    field="session_id", expected=exec_rec.get("session_id"),
                    actual=subj.get("session_id")))
            if ex.get("status") and exec_rec.get("status") and \
                    st.normalize(ex["status"]) != st.normalize(
                        exec_rec["status"]):
                issues.append(self._issue(
                    "status_mismatch", "critical",
                    "VC status differs from execution record",
                    field="status", expected=exec_rec["status"],
                    actual=ex.get("status")))
            want_in = sha256_hex(canonical_json(exec_rec.get("input")))
            want_out = sha256_hex(canonical_json(exec_rec.get("result")))
            if ex.get("input_hash") != want_in:
                hash_ok = False
                issues.append(self._issue(
                    "input_hash_mismatch", "critical",
                    "input hash does not match the stored input",
                    field="input_hash", expected=want_in,
                    actual=ex.get("input_hash")))
            if ex.get("output_hash") != want_out:
                hash_ok = False
                issues.append(self._issue(
                    "output_hash_mismatch", "critical",
                    "output hash does not match the stored result",
                    field="output_hash", expected=want_out,
                    actual=ex.get("output_hash")))
        try:
            time.strptime(doc.get("issuanceDate", ""), "%Y-%m-%dT%H:%M:%SZ")
        except ValueError:
            ts_ok = False
            issues.append(self._issue(
                "invalid_timestamp", "warning",
                "issuanceDate is not RFC3339", field="issuanceDate",
                actual=doc.get("issuanceDate")))
        struct_ok = all(doc.get(k) for k in
                        ("@context", "type", "id", "issuer", "issuanceDate"))
        if not struct_ok:
            issues.append(self._issue(
                "invalid_structure", "critical",
                "document missing required W3C fields"))
        crit = [i for i in issues if i["severity"] == "critical"]
        return {
            "metadata_consistency": not any(
                i["type"].endswith("_mismatch") for i in crit),
            "field_consistency": not any(
                i["type"] in ("execution_id_mismatch", "issuer_mismatch")
                for i in issues),
            "timestamp_validation": ts_ok,
            "hash_validation": hash_ok,
            "structural_integrity": struct_ok,
            "issues": issues,
        }

    def _security_analysis(self, rec: dict, doc: dict) -> dict:
        issues = []
        score = 100.0
        issuer = doc.get("issuer", "")
        pub = pubkey_from_did(issuer)
        did_ok = pub is not None
        if not did_ok:
            score -= 50.0
            issues.append(self._issue(
                "did_resolution_failed", "critical",
                f"cannot resolve issuer DID '{issuer}'", field="issuer"))
        sig_ok = False
        if did_ok:
            res = self.verify_document(doc)
            sig_ok = bool(res["checks"].get("signature"))
            if not sig_ok:
                score -= 40.0
                issues.append(self._issue(
                    "signature_verification_failed", "critical",
                    "Ed25519 proof does not verify against the issuer key"))
        tamper = []
        subj = doc.get("credentialSubject", {})
        if rec.get("issuer_did") and rec["issuer_did"] != doc.get("issuer"):
            tamper.append("issuer_did differs from stored record")
        if rec.get("execution_id") and \
                rec["execution_id"] != subj.get("execution_id"):
            tamper.append("execution_id differs from stored record")
        if tamper:
            score -= 20.0
            issues.append(self._issue(
                "tamper_evidence", "critical",
                "; ".join(tamper)))
        return {
            "signature_strength": "ed25519",
            "key_validation": did_ok,
            "did_authenticity": did_ok,
            "replay_protection": bool(doc.get("id")),
            "tamper_evidence": tamper,
            "security_score": max(0.0, score),
            "issues": issues,
        }

    def _compliance_checks(self, doc: dict) -> dict:
        issues = []
        ctx = doc.get("@context", [])
        w3c = "https://www.w3.org/2018/credentials/v1" in ctx
        if not w3c:
            issues.append(self._issue(
                "w3c_compliance_failure", "warning",
                "missing the W3C credentials/v1 context",
                field="@context"))
        af = "AgentFieldExecutionCredential" in doc.get("type", []) and \
            "VerifiableCredential" in doc.get("type", [])
        if not af:
            issues.append(self._issue(
                "agentfield_compliance_failure", "warning",
                "missing required credential types", field="type"))
        audit = doc.get("credentialSubject", {}).get("audit", {})
        ex = doc.get("credentialSubject", {}).get("execution", {})
        audit_ok = bool(audit.get("input_data_hash")) and \
            audit.get("input_data_hash") == ex.get("input_hash") and \
            audit.get("output_data_hash") == ex.get("output_hash")
        if not audit_ok:
            issues.append(self._issue(
                "audit_trail_inconsistency", "warning",
                "audit hashes absent or inconsistent with execution hashes",
                field="audit"))
        return {
            "w3c_compliance": w3c,
            "agentfield_standard_compliance": af,
            "audit_trail_integrity": audit_ok,
            "data_integrity_checks": bool(ex.get("input_hash")
                                          and ex.get("output_hash")),
            "issues": issues,
        }

    def _score(self, result: dict) -> float:
        score = 100.0 - 25.0 * len(result["critical_issues"]) \
            - 5.0 * len(result["warnings"])
        score = (score + result["security_analysis"]["security_score"]) / 2.0
        return max(0.0, min(100.0, score))

    def verify_execution_comprehensive(self, execution_id: str) -> dict:
        now = time.strftime("%Y-%m-%dT%H:%M:%SZ", time.gmtime())
        rec = self.storage.vc_for_execution(execution_id)
        if not rec:
            return {"valid": False, "overall_score": 0.0,
                    "critical_issues": [self._issue(
                        "vc_not_found", "critical",
                        "VC not found for execution")],
                    "warnings": [], "verification_timestamp": now}
        doc = rec["document"]
        exec_rec = self.storage.get_execution(execution_id)
        result = {"verification_timestamp": now,
                  "critical_issues": [], "warnings": []}
        result["integrity_checks"] = self._integrity_checks(rec, doc,
                                                            exec_rec)
        result["security_analysis"] = self._security_analysis(rec, doc)
        result["compliance_checks"] = self._compliance_checks(doc)
        for sec in ("integrity_checks", "security_analysis",
                    "compliance_checks"):
            for issue in result[sec]["issues"]:
                if issue["severity"] == "critical":
                    result["critical_issues"].append(issue)
                elif issue["severity"] == "warning":
                    result["warnings"].append(issue)
        result["valid"] = not result["critical_issues"]
        result["overall_score"] = self._score(result)
        return result

    def verify_chain_comprehensive(self, run_id: str) -> dict:
        """Workflow-level report: per-execution component verifications
        plus chain-integrity checks (every terminal execution carries a
        VC; issuance order consistent with execution order)."""
        now = time.strftime("%Y-%m-%dT%H:%M:%SZ", time.gmtime())
        from . import status as st
        vcs = self.storage.vcs_for_run(run_id)
        execs = self.storage.executions_by_run(run_id)
        components = {}
        for v in vcs:
            components[v["execution_id"]] = \
                self.verify_execution_comprehensive(v["execution_id"])
        chain_issues = []
        covered = set(components)
        for e in execs:
            if st.is_terminal(e.get("status", "")) and e["id"] not in covered:
                chain_issues.append(self._issue(
                    "missing_vc", "warning",
                    f"terminal execution {e['id']} has no VC",
                    component=e["id"]))
        scores = [c["overall_score"] for c in components.values()]
        result = {
            "workflow_id": run_id,
            "verification_timestamp": now,
            "components": components,
            "chain_issues": chain_issues,
            "chain_integrity": not chain_issues,
            "valid": bool(components) and all(c["valid"]
                                              for c in components.values()),
            "overall_score": round(sum(scores) / len(scores), 2)
            if scores else 0.0,
        }
        return result

    def workflow_chain(self, run_id: str) -> dict:
        vcs = self.storage.vcs_for_run(run_id)
        return {
            "workflow_id": run_id,
            "count": len(vcs),
            "credentials": [v["document"] for v in vcs],
            "all_valid": all(self.verify_document(v["document"])["valid"]
                             for v in vcs),
        }
