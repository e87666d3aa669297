"""Deterministic synthetic media provider.

Generates a reproducible library of WAV tracks (ops.audio_io.
synthetic_track) entirely in-process: the test/bench media source for an
image with no network, and the data source for end-to-end analysis
tests (reference analog: test/songs/ + provider_testing_stack).
"""

from __future__ import annotations

import io
import time
import wave
from typing import Dict, List, Optional

import numpy as np
import torch

from audiomuse_amd.mediaserver import register_provider
from audiomuse_amd.mediaserver.base import Album, Provider, Track
from audiomuse_amd.ops.audio_io import synthetic_track

_ARTISTS = ["Nova Tide", "Glass Meridian", "Cobalt Drift", "Echo Parade",
            "Silver Atlas", "Moss & Ember", "Paper Satellites", "Kite Theory"]
_GENRE_WORDS = ["Dawn", "Static", "Harbor", "Neon", "Pines", "Orbit",
                "Velvet", "Ashes"]


@register_provider("synthetic")
class SyntheticProvider(Provider):
    def __init__(self, n_albums: int = 4, tracks_per_album: int = 5,
                 seconds: float = 12.0, sr: int = 44100, seed: int = 0,
                 path_prefix: str = "/music", id_prefix: str = "",
                 **_ignored):
        """path_prefix/id_prefix simulate a second server exposing the
        SAME library under different ids and mount points — the shape
        the provider-migration wizard has to bridge."""
        self.n_albums = n_albums
        self.tracks_per_album = tracks_per_album
        self.seconds = seconds
        self.sr = sr
        self.seed = seed
        self.path_prefix = path_prefix.rstrip("/")
        self.id_prefix = id_prefix
        self._playlists: Dict[str, Dict] = {}
        self._next_pl = 1

    def list_libraries(self) -> List[Dict]:
        return [{"id": "lib1", "name": "Music", "path": self.path_prefix}]

    # -- catalogue -----------------------------------------------------

    def test_connection(self) -> bool:
        return True

    def _album(self, ai: int) -> Album:
        artist = _ARTISTS[(self.seed + ai) % len(_ARTISTS)]
        name = f"{_GENRE_WORDS[ai % len(_GENRE_WORDS)]} {ai + 1}"
        tracks = [f"{self.id_prefix}a{ai}t{ti}"
                  for ti in range(self.tracks_per_album)]
        return Album(provider_id=f"{self.id_prefix}a{ai}", name=name,
                     author=artist, track_ids=tracks)

    def get_recent_albums(self, limit: int = 0) -> List[Album]:
        albums = [self._album(i) for i in range(self.n_albums)]
        return albums[:limit] if limit else albums

    def get_tracks_from_album(self, album_id: str) -> List[Track]:
        ai = int(album_id[len(self.id_prefix) + 1:])
        album = self._album(ai)
        return [
            Track(provider_id=tid, title=f"Track {ti + 1} of {album.name}",
                  author=album.author, album=album.name,
                  duration=self.seconds,
                  file_path=f"{self.path_prefix}/{album.author}/"
                            f"{album.name}/{ti + 1}.wav")
            for ti, tid in enumerate(album.track_ids)
        ]

    def get_all_songs(self) -> List[Track]:
        out: List[Track] = []
        for a in self.get_recent_albums():
            out.extend(self.get_tracks_from_album(a.provider_id))
        return out

    def download_track(self, track_id: str) -> Optional[bytes]:
        ai, ti = track_id[len(self.id_prefix) + 1:].split("t")
        seed = self.seed * 100003 + int(ai) * 101 + int(ti)
        audio = synthetic_track(seed, seconds=self.seconds, sr=self.sr)
        pcm = (torch.clamp(audio, -1, 1) * 32767.0).to(torch.int16).numpy()
        buf = io.BytesIO()
        with wave.open(buf, "wb") as w:
            w.setnchannels(1)
            w.setsampwidth(2)
            w.setframerate(self.sr)
            w.writeframes(pcm.tobytes())
        return buf.getvalue()

    def get_lyrics(self, track_id: str) -> Optional[str]:
        if hash(track_id) % 3 == 0:
            return ("we follow the morning light across the silver water "
                    "and sing about the long road home tonight")
        return None

    # -- playlists -----------------------------------------------------

    def get_all_playlists(self) -> List[Dict]:
        return list(self._playlists.values())

    def get_playlist_track_ids(self, playlist_id: str) -> List[str]:
        p = self._playlists.get(playlist_id)
        return list(p["track_ids"]) if p else []

    def create_playlist(self, name: str, track_ids: List[str]) -> Optional[str]:
        pid = f"pl{self._next_pl}"
        self._next_pl += 1
        self._playlists[pid] = {"id": pid, "name": name,
                                "track_ids": list(track_ids)}
        return pid

    def delete_playlist(self, playlist_id: str) -> bool:
        return self._playlists.pop(playlist_id, None) is not None

    # -- listening stats ----------------------------------------------

    def get_top_played_songs(self, limit: int = 100) -> List[Track]:
        songs = self.get_all_songs()
        rng = np.random.default_rng(self.seed)
        order = rng.permutation(len(songs))
        return [songs[i] for i in order[:limit]]

    def get_last_played_time(self, track_id: str) -> Optional[float]:
        return time.time() - (hash(track_id) % (90 * 86400))
