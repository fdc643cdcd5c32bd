"""Filter a URL list against domain/extension blacklists (reference
tools/openwebtext/blacklist_urls.py, condensed: same decision rules —
blacklisted domains, non-text file extensions, malformed urls)."""

import re
import sys
from urllib.parse import urlparse

DOMAIN_BLACKLIST = {
    "500px", "aapks", "akamaihd", "amazon", "apple", "artifactfire",
    "artstation", "awwni", "bandcamp", "battleforthenet", "coinscalendar",
    "dailymotion", "deviantart", "discord", "discordapp", "dropbox", "ebay",
    "facebook", "fbcdn", "flickr", "gfycat", "giphy", "github", "google",
    "gyazo", "imgflip", "imgur", "instagram", "kickstarter", "liveleak",
    "nvidia", "photobucket", "pinimg", "pinterest", "prntscr", "puu",
    "redd", "reddit", "redditmedia", "reupp", "sli", "soundcloud",
    "spotify", "streamable", "twimg", "twitch", "twitter", "vid", "vimeo",
    "vine", "vkaao", "wikimedia", "wikipedia", "xkcd", "youtu", "youtube",
    "youtubedoubler",
}
EXTENSION_BLACKLIST = (
    ".3gp", ".7z", ".aac", ".apk", ".avi", ".bmp", ".exe", ".flac", ".flv",
    ".gif", ".gz", ".ico", ".jpeg", ".jpg", ".m4a", ".mkv", ".mov", ".mp3",
    ".mp4", ".pdf", ".png", ".rar", ".svg", ".tar", ".wav", ".webm",
    ".webp", ".wma", ".wmv", ".xz", ".zip",
)
_URL_RE = re.compile(r"^https?://", re.IGNORECASE)


def url_is_clean(url: str) -> bool:
    url = url.strip()
    if not url or " " in url or not _URL_RE.match(url):
        return False
    if url.lower().endswith(EXTENSION_BLACKLIST):
        return False
    try:
        host = urlparse(url).netloc.lower()
    except ValueError:
        return False
    labels = set(host.split(":")[0].split("."))
    return not (labels & DOMAIN_BLACKLIST)


def main():
    src, dst = sys.argv[1], sys.argv[2]
    kept = total = 0
    with open(src) as fin, open(dst, "w") as fout:
        for line in fin:
            total += 1
            if url_is_clean(line):
                fout.write(line.strip() + "\n")
                kept += 1
    print(f"kept {kept}/{total} urls")


if __name__ == "__main__":
    main()
