"""Session/checkpoint store tests (path-isolated)."""

import pytest

from adversarial_spec_amd.session import SessionState, save_checkpoint


class TestSessionState:
    def test_save_load_roundtrip(self, isolated_paths):
        s = SessionState(
            session_id="abc", spec="the spec", round=2, doc_type="prd",
            models=["local/llama-3-8b"], focus="security",
        )
        s.save()
        loaded = SessionState.load("abc")
        assert loaded.spec == "the spec"
        assert loaded.round == 2
        assert loaded.doc_type == "prd"
        assert loaded.models == ["local/llama-3-8b"]
        assert loaded.updated_at  # stamped by save

    def test_load_missing_raises(self, isolated_paths):
        with pytest.raises(FileNotFoundError):
            SessionState.load("missing")

    def test_traversal_guard_load(self, isolated_paths):
        with pytest.raises(FileNotFoundError):
            SessionState.load("../../../etc/passwd")

    def test_traversal_guard_save(self, isolated_paths):
        s = SessionState(session_id="../evil", spec="x")
        with pytest.raises(ValueError):
            s.save()

    def test_history_appends(self, isolated_paths):
        s = SessionState(session_id="h", spec="x")
        s.history.append({"round": 1, "all_agreed": False, "models": []})
        s.save()
        assert SessionState.load("h").history[0]["round"] == 1

    def test_list_sessions_sorted(self, isolated_paths):
        for sid in ("one", "two"):
            SessionState(session_id=sid, spec="x").save()
        entries = SessionState.list_sessions()
        ids = {e["session_id"] for e in entries}
        assert ids == {"one", "two"}
        # newest first
        assert entries[0]["updated_at"] >= entries[1]["updated_at"]

    def test_list_skips_corrupt(self, isolated_paths):
        from adversarial_spec_amd import session as mod

        mod.SESSIONS_DIR.mkdir(parents=True, exist_ok=True)
        (mod.SESSIONS_DIR / "bad.json").write_text("{")
        SessionState(session_id="good", spec="x").save()
        assert [e["session_id"] for e in SessionState.list_sessions()] == ["good"]


class TestCheckpoints:
    def test_without_session(self, isolated_paths, monkeypatch):
        from adversarial_spec_amd import session as mod

        monkeypatch.setattr(mod, "CHECKPOINTS_DIR", isolated_paths / "ckpt")
        p = save_checkpoint("spec body", 3)
        assert p.name == "round-3.md"
        assert p.read_text() == "spec body"

    def test_with_session(self, isolated_paths, monkeypatch):
        from adversarial_spec_amd import session as mod

        monkeypatch.setattr(mod, "CHECKPOINTS_DIR", isolated_paths / "ckpt")
        p = save_checkpoint("s", 1, "sess")
        assert p.name == "sess-round-1.md"

    def test_traversal_guard(self, isolated_paths, monkeypatch):
        from adversarial_spec_amd import session as mod

        monkeypatch.setattr(mod, "CHECKPOINTS_DIR", isolated_paths / "ckpt")
        with pytest.raises(ValueError):
            save_checkpoint("s", 1, "../../evil")


class TestMutationKillers:
    """Killers for tools/mutation_check.py survivors (session store)."""

    def test_defaults(self, isolated_paths):
        from adversarial_spec_amd.session import SessionState

        st = SessionState(session_id="k1", spec="s")
        assert st.round == 1
        assert st.preserve_intent is False

    def test_save_twice_is_idempotent(self, isolated_paths):
        """mkdir(exist_ok=True): the second save hits an existing dir."""
        from adversarial_spec_amd.session import SessionState

        st = SessionState(session_id="k2", spec="s")
        st.save()
        st.round = 2
        st.save()  # exist_ok=False mutant raises FileExistsError here
        from adversarial_spec_amd.session import SessionState as S

        assert S.load("k2").round == 2

    def test_checkpoint_twice_same_dir(self, isolated_paths):
        from adversarial_spec_amd.session import save_checkpoint

        p1 = save_checkpoint("spec v1", 1, None)
        p2 = save_checkpoint("spec v2", 2, None)
        assert p1.exists() and p2.exists()

    def test_session_file_indent_format(self, isolated_paths):
        """Session JSON is the reference's 2-space-indented format."""
        from adversarial_spec_amd import session as sess
        from adversarial_spec_amd.session import SessionState

        st = SessionState(session_id="k3", spec="s")
        st.save()
        text = (sess.SESSIONS_DIR / "k3.json").read_text()
        assert '\n  "spec"' in text
        assert '\n   "spec"' not in text

    def test_save_creates_nested_parents(self, tmp_path, monkeypatch):
        """mkdir(parents=True): the sessions/checkpoints dirs may live
        under a not-yet-created config tree."""
        from adversarial_spec_amd import session as sess
        from adversarial_spec_amd.session import SessionState, save_checkpoint

        monkeypatch.setattr(sess, "SESSIONS_DIR", tmp_path / "a" / "b" / "sessions")
        monkeypatch.setattr(sess, "CHECKPOINTS_DIR", tmp_path / "c" / "d" / "ckpt")
        SessionState(session_id="k9", spec="s").save()
        assert (sess.SESSIONS_DIR / "k9.json").exists()
        p = save_checkpoint("spec", 1, None)
        assert p.exists()
