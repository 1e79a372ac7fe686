# ---
# cmd: ["python", "-m", "modal_examples_amd", "run", "examples/10_integrations/webscraper.py"]
# ---
# # Scheduled web scraper over `.map` (webscraper.py role)
#
# The reference's scraper fans `get_links` out over pages with `.map` and
# runs on a schedule (10_integrations/webscraper.py:25-60).  Same shape on a
# no-egress node: the site being scraped is itself served by the framework
# (a small hypermedia app with interlinked pages), scrape workers fetch each
# page over HTTP and extract hrefs with html.parser, and the crawl frontier
# is deduplicated through a shared `modal.Dict`.

import modal_examples_amd as modal

app = modal.App("example-webscraper")

seen = modal.Dict.from_name("scraper-seen", create_if_missing=True)

PAGES = {
    "index": ["articles", "about"],
    "articles": ["post-1", "post-2", "index"],
    "post-1": ["articles", "post-2"],
    "post-2": ["articles"],
    "about": ["index"],
}


@app.function(scaledown_window=0.5)
@modal.wsgi_app(label="site")
def site():
    def wsgi(environ, start_response):
        page = environ.get("PATH_INFO", "/").strip("/") or "index"
        # relative hrefs so the links stay under the ingress mount prefix
        links = "".join(f'<a href="{t}">{t}</a>' for t in PAGES.get(page, []))
        body = f"<html><body><h1>{page}</h1>{links}</body></html>".encode()
        start_response("200 OK", [("Content-Type", "text/html")])
        return [body]

    return wsgi


@app.function(scaledown_window=0.5)
def get_links(url: str) -> list:
    """Fetch one page, return the absolute URLs it links to."""
    import urllib.parse
    import urllib.request
    from html.parser import HTMLParser

    class LinkParser(HTMLParser):
        def __init__(self):
            super().__init__()
            self.links = []

        def handle_starttag(self, tag, attrs):
            if tag == "a":
                for k, v in attrs:
                    if k == "href":
                        self.links.append(urllib.parse.urljoin(url, v))

    with urllib.request.urlopen(url, timeout=10) as r:
        html = r.read().decode()
    p = LinkParser()
    p.feed(html)
    return p.links


@app.local_entrypoint()
def main():
    import socket

    from modal_examples_amd.web import ingress

    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    base = ingress.serve(app, port=port)
    try:
        seen.clear()
        frontier = [base + "/site/index"]
        crawled = 0
        while frontier:
            batch, frontier = frontier, []
            fresh = [u for u in batch if seen.put_if_absent(u, True)]
            for links in get_links.map(fresh):
                crawled += 1
                frontier.extend(l for l in links if l not in seen)
        assert crawled == len(PAGES), crawled  # every page exactly once
        print(f"crawled {crawled} pages, frontier drained, "
              f"{len(PAGES)} distinct pages linked")
    finally:
        ingress.stop_serving()
