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

    def workflow_chain(self, run_id: str) -> dict:
        vcs = self.storage.vcs_for_run(run_id)
        return {
            "workflow_id": run_id,
            "count": len(vcs),
            "credentials": [v["document"] for v in vcs],
            "all_valid": all(self.verify_document(v["document"])["valid"]
                             for v in vcs),
        }
