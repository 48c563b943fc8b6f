# Network access from inside the sandbox: loopback echo client+server in
# one payload (self-contained, unlike the reference's which needs an
# external server on :9999).
import socket
import threading

srv = socket.socket(socket.AF_INET, socket.SOCK_STREAM)
srv.bind(("127.0.0.1", 0))
srv.listen(1)
port = srv.getsockname()[1]

def echo_once():
    conn, _ = srv.accept()
    with conn:
        conn.sendall(conn.recv(1024))

t = threading.Thread(target=echo_once)
t.start()
with socket.create_connection(("127.0.0.1", port)) as c:
    c.sendall(b"ping over loopback")
    print("echoed:", c.recv(1024).decode())
t.join()
srv.close()
