CREATE TABLE d (h STRING, ts TIMESTAMP TIME INDEX, v DOUBLE, PRIMARY KEY (h));
INSERT INTO d (h, ts, v) VALUES ('x', 1000, 1.0), ('x', 1000, 2.0);
SELECT h, ts, v FROM d;
ADMIN flush_table('d');
INSERT INTO d (h, ts, v) VALUES ('x', 1000, 9.0);
SELECT h, ts, v FROM d;
SELECT count(*), sum(v) FROM d;
