CREATE TABLE vf (ts TIMESTAMP TIME INDEX, h STRING PRIMARY KEY, emb VECTOR(3));
INSERT INTO vf VALUES (1000,'a','[1,0,0]'),(2000,'b','[0,1,0]'),(3000,'c','[0.7,0.7,0]');
SELECT h, vec_dim(emb) FROM vf ORDER BY h;
SELECT h, vec_cos_distance(emb, '[1,0,0]') AS d FROM vf ORDER BY d LIMIT 2;
SELECT h, vec_to_string(emb) FROM vf WHERE h = 'a';
