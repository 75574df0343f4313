#!/usr/bin/env python3
"""Image classification over gRPC — same pipeline as image_client.py
forced onto the gRPC transport (reference ships it as a separate
generated-stub program; here both transports share one implementation)."""
import sys

import image_client

if __name__ == "__main__":
    sys.argv.extend(["-i", "grpc"])
    image_client.main()
